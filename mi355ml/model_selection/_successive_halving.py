"""SuccessiveHalvingSearchCV (dask_ml/model_selection/_successive_halving.py
parity): _adapt keeps the top n//aggressiveness each rung and multiplies
their partial_fit budget by aggressiveness (Li et al.)."""

from __future__ import annotations

import math

from ._incremental import BaseIncrementalSearchCV


class SuccessiveHalvingSearchCV(BaseIncrementalSearchCV):
    def __init__(self, estimator, parameters, n_initial_parameters=10,
                 n_initial_iter=None, aggressiveness=3, test_size=0.15,
                 patience=False, tol=1e-3, max_iter=100, random_state=0,
                 verbose=False):
        super().__init__(estimator, parameters, n_initial_parameters,
                         test_size, patience, tol, max_iter, random_state,
                         verbose=verbose)
        self.aggressiveness = aggressiveness
        self.n_initial_iter = n_initial_iter

    def _initial_budget(self) -> int:
        return int(self.n_initial_iter or 1)

    def _budget(self, rung: int) -> int:
        r0 = self.n_initial_iter or 1
        return int(r0 * self.aggressiveness ** rung)

    def _adapt(self, trials, round_idx):
        eta = self.aggressiveness
        alive = sorted(trials, key=lambda t: -t.score)
        keep = max(1, int(len(alive) // eta))
        if len(alive) == 1:
            keep = 1
        survivors = alive[:keep]
        # Rung budget r_i = r0*eta^rung is the TARGET TOTAL number of
        # partial_fit calls after the rung (reference semantics), so each
        # survivor is granted r_i - pf_calls additional calls, clamped to
        # max_iter (advisor r1 medium: granting r_i as *extra* calls
        # inflated training cost ~eta/(eta-1)x).
        budget = self._budget(round_idx)
        out = {}
        for t in survivors:
            extra = min(budget, self.max_iter) - t.pf_calls
            if extra > 0:
                out[t.tid] = extra
        if len(survivors) <= 1 and not out:
            return {}
        if not out:
            return {}
        return out
