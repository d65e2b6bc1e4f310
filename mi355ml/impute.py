"""SimpleImputer (dask_ml/impute.py parity).

NaN-aware column statistics: mean via masked sums + allreduce; median via
the distributed percentile sketch; most_frequent via the
distributed value-count union (C12); constant fill is elementwise.
"""

from __future__ import annotations

import numpy as np
import torch

from .base import BaseEstimator, TransformerMixin, check_array, check_is_fitted
from .core.array import DistArray
from .ops.unique import unique


class SimpleImputer(BaseEstimator, TransformerMixin):
    def __init__(self, missing_values=float("nan"), strategy: str = "mean",
                 fill_value=None, copy: bool = True):
        self.missing_values = missing_values
        self.strategy = strategy
        self.fill_value = fill_value
        self.copy = copy

    def _mask(self, t: torch.Tensor) -> torch.Tensor:
        mv = self.missing_values
        if isinstance(mv, float) and np.isnan(mv):
            return torch.isnan(t)
        return t == mv

    def fit(self, X, y=None):
        X = check_array(X, dtype=None)
        x = X._as2d()
        dev = x.device
        F = x.shape[1]
        if self.strategy == "constant":
            fv = 0.0 if self.fill_value is None else self.fill_value
            self.statistics_ = torch.full((F,), float(fv),
                                          dtype=torch.float64, device=dev)
        elif self.strategy == "mean":
            mask = self._mask(x)
            vals = torch.where(mask, torch.zeros_like(x), x).double()
            s = vals.sum(dim=0)
            n = (~mask).double().sum(dim=0)
            buf = torch.cat([s, n])
            if X.comm.world_size > 1:
                X.comm.all_reduce(buf)
            self.statistics_ = buf[:F] / buf[F:].clamp(min=1.0)
        elif self.strategy == "median":
            from .ops.percentile import col_percentiles
            self.statistics_ = col_percentiles(X, [50.0])[0]
        elif self.strategy == "most_frequent":
            stats = torch.empty(F, dtype=torch.float64, device=dev)
            for f in range(F):
                col = x[:, f]
                col = col[~self._mask(col)]
                # global_rows is unused by unique(); local-only wrapper
                ca = DistArray(col.contiguous(), 0, 0, X.comm)
                vs, cs = unique(ca, return_counts=True)
                if vs.numel() == 0:
                    stats[f] = 0.0
                else:
                    mx = cs.max()
                    # ties -> smallest value (sklearn parity)
                    stats[f] = vs[(cs == mx).nonzero()[0, 0]].double()
            self.statistics_ = stats
        else:
            raise ValueError(f"unknown strategy {self.strategy!r}")
        self.n_features_in_ = F
        return self

    def transform(self, X):
        check_is_fitted(self, "statistics_")
        X = check_array(X, dtype=None)
        stats = self.statistics_

        def _t(t):
            two_d = t.dim() == 2
            tt = t if two_d else t[:, None]
            st = stats.to(device=tt.device, dtype=tt.dtype)
            out = torch.where(self._mask(tt), st.expand_as(tt), tt)
            return out if two_d else out[:, 0]

        return X.map_local(_t)

    def fit_transform(self, X, y=None):
        return self.fit(X).transform(X)
