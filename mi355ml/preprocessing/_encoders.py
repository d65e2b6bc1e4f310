"""OneHotEncoder / OrdinalEncoder (dask_ml/preprocessing/_encoders.py
parity — numeric category columns, dense blocks; K18 gather-scatter is a
searchsorted + scatter on device)."""

from __future__ import annotations

from typing import List, Optional

import numpy as np
import torch

from ..base import BaseEstimator, TransformerMixin, check_array, check_is_fitted
from ..core.array import DistArray
from ..ops.unique import column_uniques


class _BaseEncoder(BaseEstimator, TransformerMixin):
    def __init__(self, categories="auto", dtype=torch.float32,
                 handle_unknown: str = "error"):
        self.categories = categories
        self.dtype = dtype
        self.handle_unknown = handle_unknown

    def fit(self, X, y=None):
        X = check_array(X, dtype=None)
        if self.categories == "auto":
            self.categories_ = column_uniques(X)
        else:
            self.categories_ = [torch.as_tensor(c, device=X.device)
                                for c in self.categories]
        self.n_features_in_ = X.ncols
        return self

    def _codes(self, t: torch.Tensor) -> torch.Tensor:
        """[m, F] integer codes; -1 for unknown when ignoring."""
        cols = []
        for f, cats in enumerate(self.categories_):
            cats = cats.to(t.device)
            col = t[:, f].to(cats.dtype)
            idx = torch.searchsorted(cats, col.contiguous())
            idx = idx.clamp(max=cats.shape[0] - 1)
            ok = cats[idx] == col
            if self.handle_unknown == "error":
                if not bool(ok.all()):
                    raise ValueError("unknown categories during transform")
            else:
                idx = torch.where(ok, idx, torch.full_like(idx, -1))
            cols.append(idx)
        return torch.stack(cols, dim=1)

    def fit_transform(self, X, y=None):
        return self.fit(X).transform(X)


class OrdinalEncoder(_BaseEncoder):
    def transform(self, X):
        check_is_fitted(self, "categories_")
        X = check_array(X, dtype=None)
        dt = self.dtype
        return X.map_local(lambda t: self._codes(t).to(dt))

    def inverse_transform(self, X):
        check_is_fitted(self, "categories_")
        X = check_array(X, dtype=None)
        cats = self.categories_

        def _inv(t):
            cols = [cats[f].to(t.device)[t[:, f].long()]
                    for f in range(len(cats))]
            return torch.stack(cols, dim=1)

        return X.map_local(_inv)


class OneHotEncoder(_BaseEncoder):
    def __init__(self, categories="auto", dtype=torch.float32,
                 handle_unknown: str = "error", sparse_output: bool = False,
                 drop=None):
        super().__init__(categories, dtype, handle_unknown)
        self.sparse_output = sparse_output
        self.drop = drop  # None or "first" (collinearity guard)

    def _sizes(self):
        d1 = 1 if self.drop == "first" else 0
        return [int(c.shape[0]) - d1 for c in self.categories_]

    def transform(self, X):
        check_is_fitted(self, "categories_")
        if self.drop not in (None, "first"):
            raise ValueError("drop must be None or 'first'")
        X = check_array(X, dtype=None)
        sizes = self._sizes()
        offsets = np.concatenate([[0], np.cumsum(sizes)])
        total = int(offsets[-1])
        dt = self.dtype
        if self.sparse_output:
            return self._transform_sparse(X, offsets, total)
        d1 = 1 if self.drop == "first" else 0

        def _oh(t):
            codes = self._codes(t) - d1  # dropped category -> code -1
            out = torch.zeros(t.shape[0], total, dtype=dt, device=t.device)
            for f in range(codes.shape[1]):
                c = codes[:, f]
                valid = c >= 0
                rows = torch.arange(t.shape[0], device=t.device)[valid]
                out[rows, c[valid] + int(offsets[f])] = 1
            return out

        return X.map_local(_oh)

    def _transform_sparse(self, X, offsets, total):
        """CSR shards straight from the codes — the dense one-hot matrix is
        never materialized (dask-ml sparse_output=True parity)."""
        from ..core.sparse import SparseDistArray
        t = X._as2d()
        m, F = t.shape
        dev = t.device
        codes = self._codes(t)                                   # [m, F]
        if self.drop == "first":
            codes = codes - 1
        off = torch.as_tensor(offsets[:-1], device=dev)
        valid = codes >= 0
        nnz_per_row = valid.sum(dim=1)
        crow = torch.zeros(m + 1, dtype=torch.int64, device=dev)
        crow[1:] = nnz_per_row.cumsum(0)
        cols_full = codes + off[None, :]
        col = cols_full[valid]                # row-major order = CSR order
        vals = torch.ones(col.shape[0], dtype=self.dtype, device=dev)
        return SparseDistArray(crow, col, vals, total, X.global_rows,
                               X.row_offset, X.comm)

    def get_feature_names_out(self, input_features=None):
        names = (list(input_features) if input_features is not None
                 else [f"x{i}" for i in range(self.n_features_in_)])
        out = []
        d1 = 1 if self.drop == "first" else 0
        for f, cats in enumerate(self.categories_):
            for v in cats.cpu().tolist()[d1:]:
                out.append(f"{names[f]}_{v}")
        return np.asarray(out, dtype=object)
