"""GridSearchCV / RandomizedSearchCV (dask_ml/model_selection/_search.py
parity).

The reference's distinguishing feature is building one task graph for the
whole CV x candidate grid with shared pipeline prefixes deduplicated by
token. The eager SPMD equivalent: candidates run as
distributed fits (every rank participates in each fit over the row-sharded
folds), and for Pipeline estimators the fit_transform of each distinct
(prefix-params, fold) pair is computed ONCE and cached by token — a vec
shared by all clf-grid candidates is fitted once per fold, not once per
candidate.
"""

from __future__ import annotations

import numbers
from typing import Any, Dict, List, Optional

import numpy as np
from sklearn.base import clone
from sklearn.model_selection import ParameterGrid, ParameterSampler

from ..base import BaseEstimator, check_array
from ..core.array import DistArray
from ._split import check_cv, mask_take


def _tokenize(obj: Any) -> str:
    """Deterministic content token (the reference's dask.base.tokenize
    analog, dask/base.py:789) for params/estimators."""
    import hashlib
    import json

    def norm(o):
        if isinstance(o, dict):
            return {str(k): norm(v) for k, v in sorted(o.items())}
        if isinstance(o, (list, tuple)):
            return [norm(v) for v in o]
        if isinstance(o, (str, bool, numbers.Number)) or o is None:
            return o
        if hasattr(o, "get_params"):
            return {"__est__": type(o).__name__, **norm(o.get_params(deep=False))}
        return repr(o)

    return hashlib.sha1(
        json.dumps(norm(obj), sort_keys=True, default=repr).encode()
    ).hexdigest()


def _is_pipeline(est) -> bool:
    return hasattr(est, "steps") and isinstance(getattr(est, "steps"), list)


class _SearchBase(BaseEstimator):
    #: row*col threshold under which "auto" switches to candidate-parallel
    _SMALL_DATA_CELLS = 2_000_000

    def __init__(self, estimator, scoring=None, cv=None, refit=True,
                 return_train_score=False, candidate_parallel="auto",
                 error_score=np.nan):
        self.estimator = estimator
        self.scoring = scoring
        self.cv = cv
        self.refit = refit
        self.return_train_score = return_train_score
        # "auto": small data + world>1 -> one candidate subset per rank
        # (each rank fits its candidates on the locally-gathered data, the
        # reference's cluster-wide candidate fan-out); "ranks" forces it;
        # "none" keeps every candidate a distributed fit.
        self.candidate_parallel = candidate_parallel
        # sklearn semantics: "raise" re-raises a failing candidate fit;
        # a number is recorded as that candidate's score (with a warning)
        self.error_score = error_score

    def _candidates(self) -> List[Dict]:
        raise NotImplementedError

    def _scorers(self):
        """Normalize self.scoring into ({name: scorer}, multimetric).
        Single metric keeps the legacy name 'score' (sklearn parity)."""
        from ..metrics.scorer import check_scoring, get_scorer
        s = self.scoring
        if s is None or isinstance(s, str) or callable(s):
            return {"score": check_scoring(self.estimator, s)}, False
        if isinstance(s, (list, tuple, set)):
            return {name: get_scorer(name) for name in s}, True
        if isinstance(s, dict):
            return {name: get_scorer(sc) for name, sc in s.items()}, True
        raise ValueError(f"unsupported scoring spec {s!r}")

    def _score(self, est, Xte, yte) -> float:
        scorers, multi = self._scorers()
        if multi:
            name = self.refit if isinstance(self.refit, str) else \
                next(iter(scorers))
            return scorers[name](est, Xte, yte)
        return scorers["score"](est, Xte, yte)

    def _fit_candidate(self, params, Xtr, ytr, Xte, yte, cache, fold,
                       scorers):
        """Fit one candidate on one fold; returns ({name: test_score},
        {name: train_score} or None). A failing fit/score records
        self.error_score (sklearn semantics) unless it is 'raise'."""
        want_train = self.return_train_score
        try:
            est = clone(self.estimator)
            est.set_params(**params)
            if _is_pipeline(est) and len(est.steps) > 1:
                # pipeline-prefix dedup: fit_transform shared prefixes once
                Xtr_c, Xte_c = Xtr, Xte
                prefix_token = str(fold)
                steps = est.steps
                for name, step in steps[:-1]:
                    prefix_token = _tokenize([prefix_token, name, step])
                    hit = cache.get(prefix_token)
                    if hit is None:
                        stf = clone(step).fit(Xtr_c, ytr)
                        Xtr_n = stf.transform(Xtr_c)
                        Xte_n = stf.transform(Xte_c)
                        cache[prefix_token] = (stf, Xtr_n, Xte_n)
                    stf, Xtr_c, Xte_c = cache[prefix_token]
                final_name, final = steps[-1]
                fitted = clone(final)
                if ytr is not None:
                    fitted.fit(Xtr_c, ytr)
                else:
                    fitted.fit(Xtr_c)
                Xte_s, Xtr_s = Xte_c, Xtr_c
            else:
                fitted = est
                if ytr is not None:
                    fitted.fit(Xtr, ytr)
                else:
                    fitted.fit(Xtr)
                Xte_s, Xtr_s = Xte, Xtr
            test = {n: float(sc(fitted, Xte_s, yte))
                    for n, sc in scorers.items()}
            train = ({n: float(sc(fitted, Xtr_s, ytr))
                      for n, sc in scorers.items()} if want_train else None)
            return test, train
        except Exception:
            if self.error_score == "raise":
                raise
            import warnings
            warnings.warn(
                f"candidate {params!r} failed on fold {fold}; scoring it "
                f"error_score={self.error_score!r}", UserWarning)
            err = float(self.error_score)
            test = {n: err for n in scorers}
            train = {n: err for n in scorers} if want_train else None
            return test, train

    def _use_candidate_parallel(self, X, n_candidates: int) -> bool:
        if X.comm.world_size == 1 or n_candidates < 2:
            return False
        if self.candidate_parallel == "ranks":
            return True
        if self.candidate_parallel == "auto":
            return (X.global_rows * max(1, X.ncols)
                    <= self._SMALL_DATA_CELLS)
        return False

    def _score_grid(self, X, y, candidates, cv, n_splits, scorers):
        """scores[name][c, f] with every candidate fitted by ALL ranks."""
        C = len(candidates)
        test = {n: np.full((C, n_splits), np.nan) for n in scorers}
        train = ({n: np.full((C, n_splits), np.nan) for n in scorers}
                 if self.return_train_score else None)
        for f, (tr_mask, te_mask) in enumerate(cv.split(X, y)):
            arrays = [X] + ([y] if y is not None else [])
            tr = mask_take(arrays, tr_mask)
            te = mask_take(arrays, te_mask)
            Xtr, Xte = tr[0], te[0]
            ytr = tr[1] if y is not None else None
            yte = te[1] if y is not None else None
            cache: Dict[str, Any] = {}
            for c, params in enumerate(candidates):
                ts, tr_s = self._fit_candidate(
                    params, Xtr, ytr, Xte, yte, cache, f, scorers)
                for n in scorers:
                    test[n][c, f] = ts[n]
                    if train is not None:
                        train[n][c, f] = tr_s[n]
        return test, train

    def _score_grid_rank_parallel(self, X, y, candidates, cv, n_splits,
                                  scorers):
        """Candidates fan out round-robin across ranks; each rank fits its
        subset on the locally-gathered (small) data with a single-process
        communicator, then scores are all-gathered (the reference's
        cluster-wide candidate parallelism for sub-GPU-sized problems)."""
        from ..parallel import comm as comm_mod
        comm = X.comm
        lcomm = comm_mod.local()
        Xl = DistArray(X.gather(), X.global_rows, 0, lcomm)
        yl = (DistArray(y.gather(), y.global_rows, 0, lcomm)
              if y is not None else None)
        mine = list(range(comm.rank, len(candidates), comm.world_size))
        C = len(candidates)
        test = {n: np.full((C, n_splits), np.nan) for n in scorers}
        train = ({n: np.full((C, n_splits), np.nan) for n in scorers}
                 if self.return_train_score else None)
        for f, (tr_mask, te_mask) in enumerate(cv.split(Xl, yl)):
            arrays = [Xl] + ([yl] if yl is not None else [])
            tr = mask_take(arrays, tr_mask)
            te = mask_take(arrays, te_mask)
            Xtr, Xte = tr[0], te[0]
            ytr = tr[1] if yl is not None else None
            yte = te[1] if yl is not None else None
            cache: Dict[str, Any] = {}

            def _one(c):
                return c, self._fit_candidate(
                    candidates[c], Xtr, ytr, Xte, yte, cache, f, scorers)

            # candidate fits on the locally-gathered data are independent
            # (world-1 communicator, no collectives) — fan them out over
            # the intra-rank thread pool. The prefix cache is shared:
            # worst-case race is a duplicated prefix fit, never a wrong one
            from .. import config as _config
            nthreads = int(_config.get("search.threads") or 0)
            if nthreads > 1 and len(mine) > 1:
                from concurrent.futures import ThreadPoolExecutor
                ovs = _config.capture_overlays()

                def _one_w(c):
                    _config.install_overlays(ovs)
                    return _one(c)

                with ThreadPoolExecutor(min(nthreads, len(mine))) as pool:
                    results = list(pool.map(_one_w, mine))
            else:
                results = [_one(c) for c in mine]
            for c, (ts, tr_s) in results:
                for n in scorers:
                    test[n][c, f] = ts[n]
                    if train is not None:
                        train[n][c, f] = tr_s[n]
        payload = {c: {n: (test[n][c].tolist(),
                           train[n][c].tolist() if train is not None
                           else None) for n in scorers} for c in mine}
        for part in comm.all_gather_object(payload):
            for c, per_name in part.items():
                for n, (trow, trrow) in per_name.items():
                    test[n][c] = trow
                    if train is not None and trrow is not None:
                        train[n][c] = trrow
        return test, train

    @staticmethod
    def _rank(mean: np.ndarray) -> np.ndarray:
        """sklearn-style ranks, NaN (failed) candidates ranked last."""
        key = np.where(np.isnan(mean), -np.inf, mean)
        return ((-key).argsort(kind="stable").argsort() + 1).astype(int)

    def fit(self, X, y=None, **fit_params):
        X = check_array(X, dtype=None)
        if y is not None:
            y = check_array(y, dtype=None)
        cv = check_cv(self.cv)
        candidates = self._candidates()
        scorers, multimetric = self._scorers()
        if multimetric and self.refit is not False and \
                not (isinstance(self.refit, str) and self.refit in scorers):
            raise ValueError(
                "with multimetric scoring, refit must be False or the name "
                f"of a metric in {sorted(scorers)}")
        n_splits = cv.get_n_splits(X, y)
        if self._use_candidate_parallel(X, len(candidates)):
            test, train = self._score_grid_rank_parallel(
                X, y, candidates, cv, n_splits, scorers)
        else:
            test, train = self._score_grid(X, y, candidates, cv, n_splits,
                                           scorers)
        self.cv_results_ = {"params": candidates}
        self.multimetric_ = multimetric
        suffix = (lambda n: n) if multimetric else (lambda n: "score")
        for n in scorers:
            sn = suffix(n)
            mean = test[n].mean(axis=1)
            self.cv_results_[f"mean_test_{sn}"] = mean
            self.cv_results_[f"std_test_{sn}"] = test[n].std(axis=1)
            self.cv_results_[f"rank_test_{sn}"] = self._rank(mean)
            for f in range(n_splits):
                self.cv_results_[f"split{f}_test_{sn}"] = test[n][:, f]
            if train is not None:
                self.cv_results_[f"mean_train_{sn}"] = train[n].mean(axis=1)
                self.cv_results_[f"std_train_{sn}"] = train[n].std(axis=1)
                for f in range(n_splits):
                    self.cv_results_[f"split{f}_train_{sn}"] = train[n][:, f]
        refit_name = (self.refit if isinstance(self.refit, str)
                      else next(iter(scorers)))
        mean = self.cv_results_[f"mean_test_{suffix(refit_name)}"]
        if np.isnan(mean).all():
            raise ValueError("every candidate failed; cannot pick a best "
                             "(error_score recorded NaN for all)")
        best = int(np.nanargmax(mean))
        self.best_index_ = best
        self.best_params_ = candidates[best]
        self.best_score_ = float(mean[best])
        if self.refit is not False and self.refit is not None:
            est = clone(self.estimator)
            est.set_params(**self.best_params_)
            if y is not None:
                est.fit(X, y)
            else:
                est.fit(X)
            self.best_estimator_ = est
        return self

    def predict(self, X):
        return self.best_estimator_.predict(X)

    def predict_proba(self, X):
        return self.best_estimator_.predict_proba(X)

    def predict_log_proba(self, X):
        return self.best_estimator_.predict_log_proba(X)

    def decision_function(self, X):
        return self.best_estimator_.decision_function(X)

    def transform(self, X):
        return self.best_estimator_.transform(X)

    def inverse_transform(self, X):
        return self.best_estimator_.inverse_transform(X)

    @property
    def classes_(self):
        return self.best_estimator_.classes_

    def score(self, X, y=None):
        return self._score(self.best_estimator_, check_array(X, dtype=None),
                           check_array(y, dtype=None) if y is not None
                           else None)


class GridSearchCV(_SearchBase):
    def __init__(self, estimator, param_grid, scoring=None, cv=None,
                 refit=True, return_train_score=False,
                 candidate_parallel="auto", error_score=np.nan):
        super().__init__(estimator, scoring, cv, refit, return_train_score,
                         candidate_parallel, error_score)
        self.param_grid = param_grid

    def _candidates(self):
        return list(ParameterGrid(self.param_grid))


class RandomizedSearchCV(_SearchBase):
    def __init__(self, estimator, param_distributions, n_iter=10,
                 scoring=None, cv=None, refit=True, random_state=0,
                 return_train_score=False, candidate_parallel="auto",
                 error_score=np.nan):
        super().__init__(estimator, scoring, cv, refit, return_train_score,
                         candidate_parallel, error_score)
        self.param_distributions = param_distributions
        self.n_iter = n_iter
        self.random_state = random_state

    def _candidates(self):
        return list(ParameterSampler(self.param_distributions, self.n_iter,
                                     random_state=self.random_state))
