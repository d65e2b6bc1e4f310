"""ColumnTransformer (dask_ml.compose parity): apply different transformers
to column subsets of a DistArray and concatenate the results blockwise."""

from __future__ import annotations

from typing import List, Optional, Tuple

import torch
from sklearn.base import clone

from .base import BaseEstimator, TransformerMixin, check_array, check_is_fitted
from .core.array import DistArray


def _select(X: DistArray, cols) -> DistArray:
    cols = list(cols)
    return X.map_local(lambda t: t[:, cols].contiguous())


class ColumnTransformer(BaseEstimator, TransformerMixin):
    """transformers: list of (name, transformer, columns); remainder in
    {'drop', 'passthrough'}."""

    def __init__(self, transformers: List[Tuple], remainder: str = "drop"):
        self.transformers = transformers
        self.remainder = remainder

    def _remainder_cols(self, d: int):
        used = set()
        for _, _, cols in self.transformers:
            used.update(cols)
        return [c for c in range(d) if c not in used]

    def fit(self, X, y=None):
        X = check_array(X, dtype=None)
        self.transformers_ = []
        for name, tf, cols in self.transformers:
            sub = _select(X, cols)
            if tf == "passthrough":
                fitted = "passthrough"
            elif tf == "drop":
                fitted = "drop"
            else:
                fitted = clone(tf).fit(sub, y)
            self.transformers_.append((name, fitted, list(cols)))
        self.n_features_in_ = X.ncols
        return self

    def transform(self, X):
        check_is_fitted(self, "transformers_")
        X = check_array(X, dtype=None)
        outs = []
        for name, fitted, cols in self.transformers_:
            sub = _select(X, cols)
            if fitted == "drop":
                continue
            if fitted == "passthrough":
                outs.append(sub)
            else:
                outs.append(fitted.transform(sub))
        if self.remainder == "passthrough":
            rem = self._remainder_cols(X.ncols)
            if rem:
                outs.append(_select(X, rem))
        if not outs:
            raise ValueError("all columns dropped")
        locals_ = [o._as2d() for o in outs]
        return X.map_local(lambda t: torch.cat(
            [loc.to(t.dtype) for loc in locals_], dim=1))

    def fit_transform(self, X, y=None):
        return self.fit(X, y).transform(X)


def make_column_transformer(*transformers, remainder: str = "drop"
                            ) -> ColumnTransformer:
    named = [(f"t{i}", tf, cols) for i, (tf, cols) in enumerate(transformers)]
    return ColumnTransformer(named, remainder=remainder)
