"""PCA (dask_ml/decomposition/pca.py parity — SURVEY §2.1).

Mean-center (fused column-stats pass + one small allreduce) then SVD:
svd_solver 'full'/'tsqr' -> distributed TSQR (ops/linalg.tsqr);
'randomized' -> Halko range finder (ops/linalg.svd_compressed);
'auto' -> randomized when n_components < 0.8·ncols, else tsqr.
"""

from __future__ import annotations

from typing import Optional

import torch

from ..base import BaseEstimator, TransformerMixin, check_array, check_is_fitted
from ..core.array import DistArray
from ..ops import affine_transform
from ..ops.linalg import svd_compressed, tsqr


class PCA(BaseEstimator, TransformerMixin):
    def __init__(self, n_components: Optional[int] = None,
                 whiten: bool = False, svd_solver: str = "auto",
                 iterated_power: int = 2, random_state: int = 0):
        self.n_components = n_components
        self.whiten = whiten
        self.svd_solver = svd_solver
        self.iterated_power = iterated_power
        self.random_state = random_state

    def _solver(self, d: int, k: int) -> str:
        if self.svd_solver != "auto":
            return self.svd_solver
        return "randomized" if k < 0.8 * d else "full"

    def _fit(self, X: DistArray, need_u: bool = False):
        from ..core.colarray import ColDistArray
        if isinstance(X, ColDistArray):
            return self._fit_col(X, need_u)
        d = X.ncols
        n = X.global_rows
        k = self.n_components if self.n_components is not None else min(n, d)
        mean = X.mean(axis=0).to(X.dtype)
        ones = torch.ones(d, dtype=X.dtype, device=X.device)
        Xc = X.map_local(lambda t: affine_transform(t, ones, -mean))
        solver = self._solver(d, k)
        if solver in ("full", "tsqr"):
            U, s, Vh = tsqr(Xc, compute_svd=True, compute_q=need_u)
            Uk = (DistArray(U._as2d()[:, :k], X.global_rows, X.row_offset,
                            X.comm) if need_u else None)
            s = s[:k]
            Vh = Vh[:k]
        elif solver == "randomized":
            U, s, Vh = svd_compressed(
                Xc, k, n_power_iter=self.iterated_power,
                seed=int(self.random_state or 0), compute_u=need_u)
            Uk = U
        else:
            raise ValueError(f"unknown svd_solver {solver!r}")
        n_f = float(n)
        exp_var = (s.double() ** 2) / max(n_f - 1, 1.0)
        # total variance from the exact column moments (all components)
        _, _, m2 = Xc.moments()
        total_var = float(m2.sum() / max(n_f - 1, 1.0))
        self.mean_ = mean
        self.components_ = Vh
        self.singular_values_ = s
        self.explained_variance_ = exp_var.to(X.dtype)
        self.explained_variance_ratio_ = (
            (exp_var / max(total_var, 1e-300)).to(X.dtype))
        if k < min(n, d):
            self.noise_variance_ = (
                (total_var - float(exp_var.sum())) / (min(n, d) - k))
        else:
            self.noise_variance_ = 0.0
        self.n_components_ = k
        self.n_features_in_ = d
        self.n_samples_ = n
        return Uk, s, Vh

    def _fit_col(self, X, need_u: bool = False):
        """Wide-X feature-parallel PCA: per-column centering
        is LOCAL on a column shard; the SVD is the col-sharded randomized
        range finder (ops.linalg.svd_randomized_col)."""
        from ..core.colarray import ColDistArray, Replicated
        from ..ops.linalg import svd_randomized_col
        d = X.ncols
        n = X.global_rows
        k = self.n_components if self.n_components is not None else min(n, d)
        mean_loc = X.local.mean(dim=0).to(X.dtype)
        Xc = ColDistArray(X.local - mean_loc[None, :], d, X.col_offset,
                          X.comm)
        u, s, Vh = svd_randomized_col(
            Xc, k, n_power_iter=max(int(self.iterated_power or 0), 2),
            seed=int(self.random_state or 0), compute_u=need_u)
        n_f = float(n)
        exp_var = (s.double() ** 2) / max(n_f - 1, 1.0)
        total_var = float(
            Xc.var(axis=0).double().sum() * n_f / max(n_f - 1, 1.0))
        self.mean_ = X.mean(axis=0).to(X.dtype)   # full [d] (gathered)
        self.components_ = Vh
        self.singular_values_ = s
        self.explained_variance_ = exp_var.to(X.dtype)
        self.explained_variance_ratio_ = (
            (exp_var / max(total_var, 1e-300)).to(X.dtype))
        if k < min(n, d):
            self.noise_variance_ = (
                (total_var - float(exp_var.sum())) / (min(n, d) - k))
        else:
            self.noise_variance_ = 0.0
        self.n_components_ = k
        self.n_features_in_ = d
        self.n_samples_ = n
        Uk = Replicated(u, X.comm) if need_u else None
        return Uk, s, Vh

    def fit(self, X, y=None):
        X = check_array(X, col_ok=True)
        self._fit(X)
        return self

    def fit_transform(self, X, y=None):
        X = check_array(X, col_ok=True)
        U, s, _ = self._fit(X, need_u=True)
        if self.whiten:
            import math
            scale = math.sqrt(max(X.global_rows - 1, 1))
            return U.map_local(lambda t: t * scale)
        return U.map_local(lambda t: t * s.to(t.dtype))

    def transform(self, X):
        check_is_fitted(self, "components_")
        X = check_array(X, col_ok=True)
        V = self.components_.to(X.device)
        mean = self.mean_.to(X.device)
        from ..core.colarray import ColDistArray, Replicated
        if isinstance(X, ColDistArray):
            # feature-parallel: Σ_r (X_loc - μ_loc) V_locᵀ → one [m, k]
            # allreduce
            lo, hi = X.col_offset, X.col_offset + X.local_cols
            z = (X.local - mean[lo:hi][None, :]) @ V[:, lo:hi].to(X.dtype).t()
            if X.comm.world_size > 1:
                X.comm.all_reduce(z)
            if self.whiten:
                z = z * (1.0 / self.explained_variance_.to(
                    z.device).sqrt()).to(z.dtype)
            return Replicated(z, X.comm)
        if self.whiten:
            inv = (1.0 / self.explained_variance_.to(X.device).sqrt())

            def _t(t):
                return ((t - mean) @ V.to(t.dtype).t()) * inv.to(t.dtype)
        else:
            def _t(t):
                return (t - mean) @ V.to(t.dtype).t()
        return X.map_local(_t)

    def inverse_transform(self, X):
        check_is_fitted(self, "components_")
        X = check_array(X, dtype=None)
        V = self.components_.to(X.device)
        mean = self.mean_.to(X.device)
        if self.whiten:
            sc = self.explained_variance_.to(X.device).sqrt()

            def _t(t):
                return (t * sc.to(t.dtype)) @ V.to(t.dtype) + mean
        else:
            def _t(t):
                return t @ V.to(t.dtype) + mean
        return X.map_local(_t)
