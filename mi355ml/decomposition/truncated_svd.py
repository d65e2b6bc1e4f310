"""TruncatedSVD (dask_ml/decomposition/truncated_svd.py parity).

Same machinery as PCA without centering; algorithm 'tsqr' (exact) or
'randomized' (range finder).
"""

from __future__ import annotations

import torch

from ..base import BaseEstimator, TransformerMixin, check_array, check_is_fitted
from ..core.array import DistArray
from ..ops.linalg import svd_compressed, tsqr


class TruncatedSVD(BaseEstimator, TransformerMixin):
    def __init__(self, n_components: int = 2, algorithm: str = "tsqr",
                 n_iter: int = 5, random_state: int = 0, tol: float = 0.0):
        self.n_components = n_components
        self.algorithm = algorithm
        self.n_iter = n_iter
        self.random_state = random_state
        self.tol = tol

    def _fit(self, X: DistArray, need_u: bool = False):
        k = self.n_components
        from ..core.colarray import ColDistArray, Replicated
        if isinstance(X, ColDistArray):
            # wide-X feature-parallel path: randomized SVD
            # with col-sharded Ω/B and a small replicated Gram eigh
            from ..ops.linalg import svd_randomized_col
            u, s, vh = svd_randomized_col(
                X, k, n_power_iter=self.n_iter,
                seed=int(self.random_state or 0), compute_u=need_u)
            n = float(X.global_rows)
            self.components_ = vh          # full [k, d] (k·d is small)
            self.singular_values_ = s
            exp_var = (s.double() ** 2) / max(n - 1, 1.0)
            self.explained_variance_ = exp_var.to(X.dtype)
            total_var = float(
                X.var(axis=0).double().sum() * n / max(n - 1, 1.0))
            self.explained_variance_ratio_ = (
                exp_var / max(total_var, 1e-300)).to(X.dtype)
            self.n_features_in_ = X.ncols
            U = Replicated(u, X.comm) if need_u else None
            return U, s, vh
        if self.algorithm in ("tsqr", "full"):
            U, s, Vh = tsqr(X, compute_svd=True, compute_q=need_u)
            if need_u:
                U = DistArray(U._as2d()[:, :k], X.global_rows, X.row_offset,
                              X.comm)
            s = s[:k]
            Vh = Vh[:k]
        elif self.algorithm == "randomized":
            U, s, Vh = svd_compressed(X, k, n_power_iter=self.n_iter,
                                      seed=int(self.random_state or 0),
                                      compute_u=need_u)
        else:
            raise ValueError(f"unknown algorithm {self.algorithm!r}")
        n = float(X.global_rows)
        self.components_ = Vh
        self.singular_values_ = s
        exp_var = (s.double() ** 2) / max(n - 1, 1.0)
        self.explained_variance_ = exp_var.to(X.dtype)
        _, _, m2 = X.moments()
        total_var = float(m2.sum() / max(n - 1, 1.0))
        self.explained_variance_ratio_ = (
            exp_var / max(total_var, 1e-300)).to(X.dtype)
        self.n_features_in_ = X.ncols
        return U, s, Vh

    def fit(self, X, y=None):
        self._fit(check_array(X, col_ok=True))
        return self

    def fit_transform(self, X, y=None):
        X = check_array(X, col_ok=True)
        U, s, _ = self._fit(X, need_u=True)
        return U.map_local(lambda t: t * s.to(t.dtype))

    def transform(self, X):
        check_is_fitted(self, "components_")
        X = check_array(X, col_ok=True)
        V = self.components_.to(X.device)
        from ..core.colarray import ColDistArray, Replicated
        if isinstance(X, ColDistArray):
            # Σ_r X_loc V_locᵀ → one [m, k] allreduce
            vl = V[:, X.col_offset:X.col_offset + X.local_cols]
            z = X.local @ vl.to(X.dtype).t()
            if X.comm.world_size > 1:
                X.comm.all_reduce(z)
            return Replicated(z, X.comm)
        return X.map_local(lambda t: t @ V.to(t.dtype).t())

    def inverse_transform(self, X):
        check_is_fitted(self, "components_")
        X = check_array(X, dtype=None)
        V = self.components_.to(X.device)
        return X.map_local(lambda t: t @ V.to(t.dtype))
