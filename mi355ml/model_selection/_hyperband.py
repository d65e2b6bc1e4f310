"""HyperbandSearchCV (dask_ml/model_selection/_hyperband.py parity).

Brackets computed from (max_iter, aggressiveness=3) per Li et al. 2018;
each bracket is one SuccessiveHalving run. The reference runs brackets
concurrently on one Client; here they run back to back, each bracket's
trials fanned over the ranks.
"""

from __future__ import annotations

import math
from typing import Dict

import numpy as np

from ..base import BaseEstimator, check_array
from ._successive_halving import SuccessiveHalvingSearchCV


def _brackets(max_iter: int, eta: int):
    """[(bracket_id, n_models, first_rung_iters)] per Hyperband."""
    s_max = int(math.floor(math.log(max_iter) / math.log(eta)))
    B = (s_max + 1) * max_iter
    out = []
    for s in range(s_max, -1, -1):
        n = int(math.ceil(B / max_iter * eta ** s / (s + 1)))
        r = max(1, int(max_iter * eta ** (-s)))
        out.append((s, n, r))
    return out


class HyperbandSearchCV(BaseEstimator):
    def __init__(self, estimator, parameters, max_iter=81, aggressiveness=3,
                 test_size=0.15, patience=False, tol=1e-3, random_state=0,
                 verbose=False):
        self.estimator = estimator
        self.parameters = parameters
        self.max_iter = max_iter
        self.aggressiveness = aggressiveness
        self.test_size = test_size
        self.patience = patience
        self.tol = tol
        self.random_state = random_state
        self.verbose = verbose

    @property
    def metadata(self) -> Dict:
        bs = _brackets(self.max_iter, self.aggressiveness)
        return {
            "n_models": sum(n for _, n, _ in bs),
            "brackets": [
                {"bracket": s, "n_models": n, "first_rung_iter": r}
                for s, n, r in bs],
        }

    def fit(self, X, y=None, classes=None, **fit_params):
        bs = _brackets(self.max_iter, self.aggressiveness)
        results = []
        meta = []
        for i, (s, n, r) in enumerate(bs):
            sha = SuccessiveHalvingSearchCV(
                self.estimator, self.parameters,
                n_initial_parameters=n, n_initial_iter=r,
                aggressiveness=self.aggressiveness,
                test_size=self.test_size, patience=self.patience,
                tol=self.tol, max_iter=self.max_iter,
                random_state=int(self.random_state or 0) + i,
                verbose=self.verbose)
            sha.fit(X, y, classes=classes, **fit_params)
            results.append(sha)
            meta.append({"bracket": s,
                         "n_models": sha.metadata_["n_models"],
                         "partial_fit_calls":
                             sha.metadata_["partial_fit_calls"]})
        best = max(results, key=lambda r: r.best_score_)
        self.best_estimator_ = best.best_estimator_
        self.best_params_ = best.best_params_
        self.best_score_ = best.best_score_
        self.metadata_ = {"n_models": sum(m["n_models"] for m in meta),
                          "partial_fit_calls":
                              sum(m["partial_fit_calls"] for m in meta),
                          "brackets": meta}
        params, scores, pfs = [], [], []
        for r in results:
            params.extend(r.cv_results_["params"])
            scores.extend(r.cv_results_["test_score"])
            pfs.extend(r.cv_results_["partial_fit_calls"])
        self.cv_results_ = {"params": params,
                            "test_score": np.asarray(scores),
                            "partial_fit_calls": np.asarray(pfs)}
        self.history_ = sum((r.history_ for r in results), [])
        return self

    def predict(self, X):
        from ..wrappers import _blockwise_method
        return _blockwise_method(self.best_estimator_, "predict",
                                 check_array(X, dtype=None))

    def score(self, X, y):
        from ..metrics import accuracy_score, r2_score
        yp = self.predict(X)
        y = check_array(y, dtype=None)
        if getattr(self.best_estimator_, "_estimator_type", "") == "regressor":
            return r2_score(y, yp)
        return accuracy_score(y, yp)
