"""LabelEncoder (dask_ml/preprocessing/label.py parity).

classes_ via distributed unique (ops/unique.py — one ragged all-gather
instead of the reference's tree-reduce); transform = searchsorted codes.
"""

from __future__ import annotations

import numpy as np
import torch

from ..base import BaseEstimator, TransformerMixin, check_array, check_is_fitted
from ..core.array import DistArray
from ..ops.unique import unique


def _categorical_parts(y):
    """pandas-categorical fast path detection (the reference's
    use_categorical branch: classes are KNOWN from the dtype, no data
    pass). Returns a list of Series or None."""
    import pandas as pd
    if isinstance(y, pd.Series):
        return [y] if isinstance(y.dtype, pd.CategoricalDtype) else None
    from ..core.bag import DistList
    if isinstance(y, DistList):
        parts = [p for p in y.local if p is not None and len(p)]
        if parts and all(isinstance(p, pd.Series)
                         and isinstance(p.dtype, pd.CategoricalDtype)
                         for p in parts):
            return parts
    return None


class LabelEncoder(BaseEstimator, TransformerMixin):
    def __init__(self, use_categorical: bool = True):
        self.use_categorical = use_categorical

    def fit(self, y):
        parts = _categorical_parts(y) if self.use_categorical else None
        if parts is not None:
            cats = parts[0].cat.categories
            for p in parts[1:]:
                if not p.cat.categories.equals(cats):
                    raise ValueError(
                        "categorical partitions have mismatched categories;"
                        " run Categorizer first for a global union")
            self.classes_ = np.asarray(cats)
            self._categorical_ = True
            return self
        y = check_array(y, dtype=None)
        self.classes_ = unique(y)
        self._categorical_ = False
        return self

    def fit_transform(self, y):
        return self.fit(y).transform(y)

    def transform(self, y):
        check_is_fitted(self, "classes_")
        if getattr(self, "_categorical_", False):
            import pandas as pd
            from ..core.bag import DistList

            def _codes(s: "pd.Series"):
                cat = s.astype(pd.CategoricalDtype(self.classes_))
                codes = cat.cat.codes.to_numpy()
                if (codes < 0).any():
                    raise ValueError("y contains previously unseen labels")
                return codes

            if isinstance(y, pd.Series):
                return _codes(y)
            if isinstance(y, DistList):
                return y.map(lambda s: _codes(s) if s is not None and len(s)
                             else np.empty(0, dtype=np.int8))
            raise TypeError("fitted on categorical input; transform "
                            "expects a pandas Series or DistList of them")
        y = check_array(y, dtype=None)
        classes = self.classes_.to(y.device)

        def _t(t):
            idx = torch.searchsorted(classes, t.reshape(-1).to(classes.dtype))
            idx = idx.clamp(max=classes.shape[0] - 1)
            ok = classes[idx] == t.reshape(-1).to(classes.dtype)
            if not bool(ok.all()):
                raise ValueError("y contains previously unseen labels")
            return idx

        return y.map_local(_t)

    def inverse_transform(self, y):
        check_is_fitted(self, "classes_")
        if getattr(self, "_categorical_", False):
            import pandas as pd
            codes = np.asarray(y)
            return pd.Series(pd.Categorical.from_codes(
                codes, categories=self.classes_))
        y = check_array(y, dtype=None)
        classes = self.classes_.to(y.device)
        return y.map_local(lambda t: classes[t.reshape(-1).long()])
