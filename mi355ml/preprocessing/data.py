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


class RobustScaler(BaseEstimator, TransformerMixin):
    """Center by the median, scale by the IQR — distributed approximate
    percentiles (ops/percentile.py, SURVEY A.5 merge math)."""

    def __init__(self, with_centering: bool = True, with_scaling: bool = True,
                 quantile_range=(25.0, 75.0), copy: bool = True,
                 method: str = "default"):
        self.with_centering = with_centering
        self.with_scaling = with_scaling
        self.quantile_range = quantile_range
        self.copy = copy
        self.method = method  # "default" (curve merge) or "tdigest"

    def fit(self, X, y=None):
        from ..ops.percentile import col_percentiles
        X = check_array(X)
        self.n_features_in_ = X.ncols
        lo, hi = self.quantile_range
        qs = col_percentiles(X, [lo, 50.0, hi], method=self.method)
        dt = X.dtype
        self.center_ = qs[1].to(dt) if self.with_centering else None
        if self.with_scaling:
            self.scale_ = _handle_zeros_in_scale((qs[2] - qs[0]).to(dt))
        else:
            self.scale_ = None
        return self

    def transform(self, X):
        check_is_fitted(self, "n_features_in_")
        X = check_array(X)
        dev, dt = X.device, X.dtype
        ones = torch.ones(X.ncols, dtype=dt, device=dev)
        zeros = torch.zeros(X.ncols, dtype=dt, device=dev)
        inv = (1.0 / self.scale_.to(dev)) if self.with_scaling else ones
        c = self.center_.to(dev) if self.with_centering else zeros
        return X.map_local(
            lambda t: affine_transform(t, inv.to(dt), (-c * inv).to(dt)))

    def inverse_transform(self, X):
        check_is_fitted(self, "n_features_in_")
        X = check_array(X)
        dev, dt = X.device, X.dtype
        ones = torch.ones(X.ncols, dtype=dt, device=dev)
        zeros = torch.zeros(X.ncols, dtype=dt, device=dev)
        sc = self.scale_.to(dev) if self.with_scaling else ones
        c = self.center_.to(dev) if self.with_centering else zeros
        return X.map_local(lambda t: affine_transform(t, sc.to(dt), c.to(dt)))

    def fit_transform(self, X, y=None):
        return self.fit(X).transform(X)


class QuantileTransformer(BaseEstimator, TransformerMixin):
    """Map each column through its empirical CDF (approximate distributed
    quantiles, like the reference's per-chunk-sample approximation)."""

    def __init__(self, n_quantiles: int = 1000,
                 output_distribution: str = "uniform",
                 subsample: int = 100_000, random_state=None,
                 copy: bool = True, method: str = "default"):
        self.n_quantiles = n_quantiles
        self.output_distribution = output_distribution
        self.subsample = subsample
        self.random_state = random_state
        self.copy = copy
        self.method = method  # "default" (curve merge) or "tdigest"

    def fit(self, X, y=None):
        from ..ops.percentile import col_percentiles
        X = check_array(X)
        nq = min(self.n_quantiles, max(2, X.global_rows))
        refs = torch.linspace(0, 100, nq, dtype=torch.float64)
        self.n_quantiles_ = nq
        self.references_ = (refs / 100.0)
        self.quantiles_ = col_percentiles(X, refs.tolist(),
                                          method=self.method)  # [nq, F] f64
        self.n_features_in_ = X.ncols
        return self

    def _map(self, t: torch.Tensor, forward: bool) -> torch.Tensor:
        from ..ops.percentile import _interp_batched
        dev = t.device
        q = self.quantiles_.to(dev)                       # [nq, F]
        refs = self.references_.to(dev)                   # [nq]
        F = t.shape[1]
        xq = t.t().double().contiguous()                  # [F, m]
        if forward:
            r = _interp_batched(xq, q.t().contiguous(),
                                refs.expand(F, -1).contiguous())
        else:
            r = _interp_batched(xq, refs.expand(F, -1).contiguous(),
                                q.t().contiguous())
        return r.t().to(t.dtype).contiguous()

    def transform(self, X):
        check_is_fitted(self, "quantiles_")
        X = check_array(X)
        out = X.map_local(lambda t: self._map(t, forward=True))
        if self.output_distribution == "normal":
            import math
            eps = 1e-7

            def _ppf(t):
                return (torch.erfinv((t.double().clamp(eps, 1 - eps)) * 2 - 1)
                        * math.sqrt(2.0)).to(t.dtype)

            out = out.map_local(_ppf)
        return out

    def inverse_transform(self, X):
        check_is_fitted(self, "quantiles_")
        X = check_array(X)
        if self.output_distribution == "normal":
            import math

            def _cdf(t):
                return (0.5 * (1 + torch.erf(t.double() / math.sqrt(2.0)))
                        ).to(t.dtype)

            X = X.map_local(_cdf)
        return X.map_local(lambda t: self._map(t, forward=False))

    def fit_transform(self, X, y=None):
        return self.fit(X).transform(X)


class Normalizer(BaseEstimator, TransformerMixin):
    """Row-wise normalization (stateless blockwise)."""

    def __init__(self, norm: str = "l2", copy: bool = True):
        self.norm = norm
        self.copy = copy

    def fit(self, X, y=None):
        X = check_array(X)
        self.n_features_in_ = X.ncols
        return self

    def transform(self, X):
        X = check_array(X)

        def _n(t):
            if self.norm == "l2":
                s = t.norm(dim=1, keepdim=True)
            elif self.norm == "l1":
                s = t.abs().sum(dim=1, keepdim=True)
            elif self.norm == "max":
                s = t.abs().amax(dim=1, keepdim=True)
            else:
                raise ValueError(self.norm)
            return t / s.clamp(min=1e-12)

        return X.map_local(_n)

    def fit_transform(self, X, y=None):
        return self.fit(X).transform(X)


class PolynomialFeatures(BaseEstimator, TransformerMixin):
    """Arbitrary-degree polynomial feature expansion (+interaction_only),
    blockwise on device (the reference maps sklearn's transformer per
    block); column order matches sklearn's combinations ordering."""

    def __init__(self, degree: int = 2, interaction_only: bool = False,
                 include_bias: bool = True):
        self.degree = degree
        self.interaction_only = interaction_only
        self.include_bias = include_bias

    def _combos(self, d: int):
        import itertools
        combos = []
        if self.include_bias:
            combos.append(())
        for deg in range(1, self.degree + 1):
            it = (itertools.combinations(range(d), deg)
                  if self.interaction_only
                  else itertools.combinations_with_replacement(range(d), deg))
            combos.extend(it)
        return combos

    def fit(self, X, y=None):
        X = check_array(X)
        d = X.ncols
        if self.degree < 0:
            raise ValueError("degree must be >= 0")
        self._combos_ = self._combos(d)
        self.n_features_in_ = d
        self.n_output_features_ = len(self._combos_)
        if self.n_output_features_ > 100_000:
            raise ValueError(
                f"degree={self.degree} over {d} features expands to "
                f"{self.n_output_features_} columns; refusing (>1e5)")
        return self

    def transform(self, X):
        check_is_fitted(self, "n_output_features_")
        X = check_array(X)
        combos = self._combos_

        def _p(t):
            cols = []
            for c in combos:
                if len(c) == 0:
                    cols.append(torch.ones(t.shape[0], dtype=t.dtype,
                                           device=t.device))
                else:
                    v = t[:, c[0]].clone()
                    for j in c[1:]:
                        v = v * t[:, j]
                    cols.append(v)
            return torch.stack(cols, dim=1)

        return X.map_local(_p)

    def fit_transform(self, X, y=None):
        return self.fit(X).transform(X)

    def get_feature_names_out(self, input_features=None):
        import numpy as _np
        names = (list(input_features) if input_features is not None
                 else [f"x{i}" for i in range(self.n_features_in_)])
        out = []
        for c in self._combos_:
            if len(c) == 0:
                out.append("1")
            else:
                from collections import Counter
                parts = []
                for i, p in sorted(Counter(c).items()):
                    parts.append(names[i] if p == 1 else f"{names[i]}^{p}")
                out.append(" ".join(parts))
        return _np.asarray(out, dtype=object)
