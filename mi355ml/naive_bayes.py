"""GaussianNB (dask_ml/naive_bayes.py parity).

Per-class mean/variance via masked per-shard reductions + ONE fused
allreduce (the reference's masked dask reductions); predict is
blockwise log-likelihood.
"""

from __future__ import annotations

import math

import torch

from .base import BaseEstimator, ClassifierMixin, check_X_y, check_array, check_is_fitted
from .metrics import accuracy_score
from .ops.unique import unique


class GaussianNB(BaseEstimator, ClassifierMixin):
    def __init__(self, priors=None, var_smoothing: float = 1e-9):
        self.priors = priors
        self.var_smoothing = var_smoothing

    def fit(self, X, y):
        X, y = check_X_y(X, y, y_dtype=None)
        classes = unique(y)
        k = classes.shape[0]
        x = X._as2d()
        d = x.shape[1]
        dev = x.device
        # per-class n, sum, sumsq in one fused buffer + one allreduce
        buf = torch.zeros(k * (2 * d + 1), dtype=torch.float64, device=dev)
        yl = y.local.reshape(-1)
        for i in range(k):
            mask = yl == classes[i]
            xc = x[mask].double()
            n = float(xc.shape[0])
            lo = i * (2 * d + 1)
            buf[lo] = n
            if n:
                buf[lo + 1:lo + 1 + d] = xc.sum(dim=0)
                buf[lo + 1 + d:lo + 1 + 2 * d] = (xc * xc).sum(dim=0)
        if X.comm.world_size > 1:
            X.comm.all_reduce(buf)
        counts = torch.empty(k, dtype=torch.float64, device=dev)
        theta = torch.empty(k, d, dtype=torch.float64, device=dev)
        var = torch.empty(k, d, dtype=torch.float64, device=dev)
        for i in range(k):
            lo = i * (2 * d + 1)
            n = buf[lo].clamp(min=1.0)
            s = buf[lo + 1:lo + 1 + d]
            ss = buf[lo + 1 + d:lo + 1 + 2 * d]
            counts[i] = buf[lo]
            theta[i] = s / n
            var[i] = (ss / n - (s / n) ** 2).clamp(min=0.0)
        eps = self.var_smoothing * float(var.max())
        self.classes_ = classes
        self.class_count_ = counts
        total = counts.sum()
        if self.priors is not None:
            self.class_prior_ = torch.as_tensor(self.priors,
                                                dtype=torch.float64,
                                                device=dev)
        else:
            self.class_prior_ = counts / total
        self.theta_ = theta
        self.var_ = var + eps
        self.n_features_in_ = d
        return self

    def _joint_log_likelihood(self, t: torch.Tensor) -> torch.Tensor:
        theta = self.theta_.to(t.device)
        var = self.var_.to(t.device)
        prior = self.class_prior_.to(t.device)
        x = t.double()
        jll = []
        for i in range(theta.shape[0]):
            lp = (-0.5 * (math.log(2 * math.pi) + var[i].log())
                  - 0.5 * (x - theta[i]) ** 2 / var[i]).sum(dim=1)
            jll.append(lp + prior[i].clamp(min=1e-300).log())
        return torch.stack(jll, dim=1)

    def predict(self, X):
        check_is_fitted(self, "theta_")
        X = check_array(X)
        classes = self.classes_

        def _p(t):
            jll = self._joint_log_likelihood(t)
            return classes.to(t.device)[jll.argmax(dim=1)]

        return X.map_local(_p)

    def predict_proba(self, X):
        check_is_fitted(self, "theta_")
        X = check_array(X)

        def _p(t):
            jll = self._joint_log_likelihood(t)
            return torch.softmax(jll, dim=1).to(t.dtype)

        return X.map_local(_p)

    def predict_log_proba(self, X):
        check_is_fitted(self, "theta_")
        X = check_array(X)

        def _p(t):
            jll = self._joint_log_likelihood(t)
            return torch.log_softmax(jll, dim=1).to(t.dtype)

        return X.map_local(_p)

    def score(self, X, y):
        return accuracy_score(check_array(y, dtype=None), self.predict(X))
