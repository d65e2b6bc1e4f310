"""Scalers (dask_ml/preprocessing/data.py parity).

fit: one fused stats pass per shard + one tiny collective (C1).
transform: one fused affine kernel pass (K9), never the reference's chain of
elemwise temporaries.
"""

from __future__ import annotations

import torch

from ..base import BaseEstimator, TransformerMixin, check_array, check_is_fitted
from ..core.array import DistArray
from ..ops import affine_transform


def _handle_zeros_in_scale(scale: torch.Tensor) -> torch.Tensor:
    """sklearn parity: zero scales transform to identity
    (sklearn/preprocessing/_data.py _handle_zeros_in_scale)."""
    out = scale.clone()
    out[out == 0.0] = 1.0
    return out


class StandardScaler(BaseEstimator, TransformerMixin):
    """(x - mean) / std, columnwise, distributed exact (Chan moment merge)."""

    def __init__(self, copy: bool = True, with_mean: bool = True,
                 with_std: bool = True):
        self.copy = copy
        self.with_mean = with_mean
        self.with_std = with_std

    def fit(self, X, y=None):
        X = check_array(X)
        n, mean, m2 = X.moments()
        self.n_samples_seen_ = int(n)
        self.n_features_in_ = X.ncols
        dt = X.dtype
        self.mean_ = mean.to(dt) if self.with_mean else None
        if self.with_std:
            var = (m2 / max(n, 1.0)).to(dt)
            self.var_ = var
            self.scale_ = _handle_zeros_in_scale(var.sqrt())
        else:
            self.var_ = None
            self.scale_ = None
        return self

    def transform(self, X, copy=None):
        check_is_fitted(self, "n_features_in_")
        X = check_array(X)
        dev, dt = X.device, X.dtype
        ones = torch.ones(X.ncols, dtype=dt, device=dev)
        zeros = torch.zeros(X.ncols, dtype=dt, device=dev)
        inv = (1.0 / self.scale_.to(dev)) if self.with_std else ones
        mu = self.mean_.to(dev) if self.with_mean else zeros
        # (x - mu) * inv  ==  x * inv + (-mu * inv): one fused pass
        return X.map_local(
            lambda t: affine_transform(t, inv.to(dt), (-mu * inv).to(dt)))

    def inverse_transform(self, X, copy=None):
        check_is_fitted(self, "n_features_in_")
        X = check_array(X)
        dev, dt = X.device, X.dtype
        ones = torch.ones(X.ncols, dtype=dt, device=dev)
        zeros = torch.zeros(X.ncols, dtype=dt, device=dev)
        sc = self.scale_.to(dev) if self.with_std else ones
        mu = self.mean_.to(dev) if self.with_mean else zeros
        return X.map_local(lambda t: affine_transform(t, sc.to(dt), mu.to(dt)))

    def fit_transform(self, X, y=None):
        return self.fit(X).transform(X)


class MinMaxScaler(BaseEstimator, TransformerMixin):
    """Scale columns to [feature_range] via distributed min/max (C1)."""

    def __init__(self, feature_range=(0.0, 1.0), copy: bool = True):
        self.feature_range = feature_range
        self.copy = copy

    def fit(self, X, y=None):
        X = check_array(X)
        self.n_features_in_ = X.ncols
        mn = X.min(axis=0)
        mx = X.max(axis=0)
        lo, hi = self.feature_range
        self.data_min_ = mn
        self.data_max_ = mx
        self.data_range_ = mx - mn
        self.scale_ = (hi - lo) / _handle_zeros_in_scale(self.data_range_)
        self.min_ = lo - mn * self.scale_
        return self

    def transform(self, X):
        check_is_fitted(self, "n_features_in_")
        X = check_array(X)
        dev, dt = X.device, X.dtype
        a = self.scale_.to(device=dev, dtype=dt)
        b = self.min_.to(device=dev, dtype=dt)
        return X.map_local(lambda t: affine_transform(t, a, b))

    def inverse_transform(self, X):
        check_is_fitted(self, "n_features_in_")
        X = check_array(X)
        dev, dt = X.device, X.dtype
        inv = (1.0 / self.scale_).to(device=dev, dtype=dt)
        b = (-self.min_ / self.scale_).to(device=dev, dtype=dt)
        return X.map_local(lambda t: affine_transform(t, inv, b))

    def fit_transform(self, X, y=None):
        return self.fit(X).transform(X)
