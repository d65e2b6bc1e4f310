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


class LabelEncoder(BaseEstimator, TransformerMixin):
    def fit(self, y):
        y = check_array(y, dtype=None)
        self.classes_ = unique(y)
        return self

    def fit_transform(self, y):
        return self.fit(y).transform(y)

    def transform(self, y):
        check_is_fitted(self, "classes_")
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
        y = check_array(y, dtype=None)
        classes = self.classes_.to(y.device)
        return y.map_local(lambda t: classes[t.reshape(-1).long()])
