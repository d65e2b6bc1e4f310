"""GLM estimators (dask_ml/linear_model/glm.py parity — SURVEY §2.1).

LinearRegression / LogisticRegression / PoissonRegression with
solver ∈ {lbfgs, admm, gradient_descent, newton, proximal_grad}
(mi355ml/glm/algorithms.py). C is the inverse regularization strength
(sklearn convention): λ = 1/C. The intercept is never penalized (instead of
the reference's add-intercept-column trick the fused kernel carries the bias
natively, so no X copy is materialized).
"""

from __future__ import annotations

from typing import Optional

import numpy as np
import torch

from ..base import BaseEstimator, check_X_y, check_array, check_is_fitted, as_result
from ..core.array import DistArray
from ..glm.algorithms import SOLVERS
from ..glm.families import get_family
from ..metrics import accuracy_score, r2_score


class _GLMBase(BaseEstimator):
    family: str = "normal"

    def __init__(self, penalty="l2", C=1.0, fit_intercept=True,
                 solver="lbfgs", max_iter=100, tol=1e-4, random_state=None,
                 eta0=0.5):
        self.penalty = penalty
        self.C = C
        self.fit_intercept = fit_intercept
        self.solver = solver
        self.max_iter = max_iter
        self.tol = tol
        self.random_state = random_state
        self.eta0 = eta0

    def fit(self, X, y):
        X, y = check_X_y(X, y, col_ok=True)
        if self.solver not in SOLVERS:
            raise ValueError(f"unknown solver {self.solver!r}; "
                             f"choose from {sorted(SOLVERS)}")
        lam = 1.0 / self.C if self.C not in (None, 0) else 0.0
        no_pen = self.penalty in (None, "none")
        kw = dict(family=self.family, max_iter=self.max_iter,
                  fit_intercept=self.fit_intercept)
        if self.solver == "lbfgs":
            if not (no_pen or self.penalty == "l2"):
                raise ValueError("lbfgs supports only l2/none penalty")
            w, b, nit = SOLVERS["lbfgs"](
                X, y, lamduh=0.0 if no_pen else lam, tol=self.tol, **kw)
        elif self.solver == "admm":
            w, b, nit = SOLVERS["admm"](
                X, y, regularizer="l2" if no_pen else self.penalty,
                lamduh=0.0 if no_pen else lam, **kw)
        elif self.solver == "proximal_grad":
            w, b, nit = SOLVERS["proximal_grad"](
                X, y, regularizer="l2" if no_pen else self.penalty,
                lamduh=0.0 if no_pen else lam, tol=self.tol, **kw)
        else:
            w, b, nit = SOLVERS[self.solver](X, y, tol=self.tol, **kw)
        self.coef_ = w
        self.intercept_ = float(b)
        self.n_iter_ = nit
        self.n_features_in_ = X.ncols
        return self

    def partial_fit(self, X, y, classes=None):
        """One distributed SGD step on this batch (fused kernel + one
        allreduce) — the native-estimator analog of sklearn partial_fit,
        usable under wrappers.Incremental and the Hyperband family."""
        import math
        from ..glm.algorithms import _dist_value_grad
        X, y = check_X_y(X, y, col_ok=True)
        d = X.ncols
        dev = X.device
        if not hasattr(self, "coef_") or self.coef_ is None:
            self.coef_ = torch.zeros(d, dtype=torch.float32, device=dev)
            self.intercept_ = 0.0
            self._t = 0
            self.n_features_in_ = d
        w = self.coef_.to(dev)
        loss, gw, gb = _dist_value_grad(X, y, w, self.intercept_,
                                        self.family)
        n = max(X.global_rows, 1)
        lam = (1.0 / self.C) if (self.C and self.penalty == "l2") else 0.0
        self._t += 1
        eta = self.eta0 / math.sqrt(self._t)
        step = gw / n + lam * w.double() / n
        self.coef_ = (w.double() - eta * step).float()
        if self.fit_intercept:
            self.intercept_ = float(self.intercept_ - eta * gb / n)
        return self

    def _linear(self, X) -> DistArray:
        check_is_fitted(self, "coef_")
        X = check_array(X, col_ok=True)
        w = self.coef_.to(X.device)
        b = self.intercept_
        from ..core.sparse import SparseDistArray
        from ..core.colarray import ColDistArray, Replicated
        if isinstance(X, SparseDistArray):
            return X.dot(w).map_local(lambda z: z + b)
        if isinstance(X, ColDistArray):
            # feature-parallel: partial GEMV + one [m] allreduce; the
            # replicated result row-shards on contact with metrics
            return Replicated(X.dot(w.to(X.dtype)) + b, X.comm)
        return X.map_local(lambda t: t @ w.to(t.dtype) + b)

    def predict(self, X):
        fam = get_family(self.family)
        return self._linear(X).map_local(fam.predict)


class LogisticRegression(_GLMBase):
    family = "logistic"

    def predict(self, X):
        return self.predict_proba(X).map_local(
            lambda p: (p[:, 1] > 0.5).to(torch.float32))

    def predict_proba(self, X):
        z = self._linear(X)

        def _proba(t: torch.Tensor) -> torch.Tensor:
            p = torch.sigmoid(t)
            return torch.stack([1.0 - p, p], dim=1)

        return z.map_local(_proba)

    def decision_function(self, X):
        return self._linear(X)

    def score(self, X, y):
        return accuracy_score(check_array(y, dtype=None), self.predict(X))


class LinearRegression(_GLMBase):
    family = "normal"

    def score(self, X, y):
        return r2_score(check_array(y, dtype=None), self.predict(X))


class PoissonRegression(_GLMBase):
    family = "poisson"

    def score(self, X, y):
        return r2_score(check_array(y, dtype=None), self.predict(X))
