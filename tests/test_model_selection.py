"""model_selection: splits, searches, wrappers, Hyperband."""

import numpy as np
import pytest
from sklearn.linear_model import SGDClassifier
from sklearn.pipeline import Pipeline

from mi355ml.core.array import from_numpy
from mi355ml.linear_model import LogisticRegression
from mi355ml.model_selection import (GridSearchCV, HyperbandSearchCV,
                                     IncrementalSearchCV, KFold,
                                     RandomizedSearchCV,
                                     SuccessiveHalvingSearchCV,
                                     train_test_split)
from mi355ml.preprocessing import StandardScaler
from mi355ml.wrappers import Incremental, ParallelPostFit


def _cls(n=3000, f=6, seed=0):
    rng = np.random.default_rng(seed)
    X = rng.standard_normal((n, f)).astype(np.float32)
    w = rng.standard_normal(f) * 2
    p = 1 / (1 + np.exp(-(X @ w)))
    y = (rng.uniform(size=n) < p).astype(np.float32)
    return X, y


def test_train_test_split():
    X, y = _cls()
    Xtr, Xte, ytr, yte = train_test_split(from_numpy(X), from_numpy(y),
                                          test_size=0.3, random_state=0)
    assert Xtr.global_rows + Xte.global_rows == X.shape[0]
    assert ytr.global_rows == Xtr.global_rows
    frac = Xte.global_rows / X.shape[0]
    assert 0.25 < frac < 0.35


def test_kfold():
    X, y = _cls(n=1000)
    Xd = from_numpy(X)
    kf = KFold(n_splits=4)
    sizes = []
    for tr, te in kf.split(Xd):
        assert bool((tr ^ te).all())
        sizes.append(int(te.sum()))
    assert sum(sizes) == 1000


def test_grid_search_logreg():
    X, y = _cls()
    gs = GridSearchCV(LogisticRegression(solver="lbfgs", max_iter=50),
                      {"C": [0.01, 1.0, 100.0]}, cv=3)
    gs.fit(from_numpy(X), from_numpy(y))
    assert gs.best_params_["C"] in (0.01, 1.0, 100.0)
    assert gs.best_score_ > 0.7
    assert len(gs.cv_results_["mean_test_score"]) == 3
    pred = gs.predict(from_numpy(X)).gather().cpu().numpy()
    assert pred.shape == (X.shape[0],)


def test_grid_search_pipeline_prefix_dedup():
    X, y = _cls(n=1500)
    calls = {"n": 0}

    class CountingScaler(StandardScaler):
        def fit(self, X, y=None):
            calls["n"] += 1
            return super().fit(X, y)

    pipe = Pipeline([("sc", CountingScaler()),
                     ("clf", LogisticRegression(max_iter=30))])
    gs = GridSearchCV(pipe, {"clf__C": [0.1, 1.0, 10.0]}, cv=2)
    gs.fit(from_numpy(X), from_numpy(y))
    # the scaler prefix is shared by all 3 candidates: fitted once per fold
    # (2) + once by the refit on the full data — NOT once per candidate (7)
    assert calls["n"] == 3, calls
    assert gs.best_score_ > 0.6


def test_randomized_search():
    X, y = _cls(n=1200, seed=2)
    rs = RandomizedSearchCV(LogisticRegression(max_iter=30),
                            {"C": [0.01, 0.1, 1.0, 10.0, 100.0]},
                            n_iter=3, cv=2, random_state=0)
    rs.fit(from_numpy(X), from_numpy(y))
    assert len(rs.cv_results_["params"]) == 3


def test_parallel_post_fit():
    X, y = _cls(n=1000, seed=3)
    ppf = ParallelPostFit(SGDClassifier(loss="log_loss", max_iter=20,
                                        random_state=0))
    ppf.fit(from_numpy(X), from_numpy(y))
    pred = ppf.predict(from_numpy(X)).gather().cpu().numpy()
    assert pred.shape == (1000,)
    acc = ppf.score(from_numpy(X), from_numpy(y))
    assert acc > 0.6
    proba = ppf.predict_proba(from_numpy(X)).gather().cpu().numpy()
    assert proba.shape == (1000, 2)


def test_incremental_wrapper():
    X, y = _cls(n=2000, seed=4)
    inc = Incremental(SGDClassifier(loss="log_loss", random_state=0),
                      shuffle_blocks=False)
    inc.fit(from_numpy(X), from_numpy(y), epochs=3)
    acc = inc.score(from_numpy(X), from_numpy(y))
    assert acc > 0.7


def test_incremental_search():
    X, y = _cls(n=2000, seed=5)
    search = IncrementalSearchCV(
        SGDClassifier(loss="log_loss", random_state=0),
        {"alpha": [1e-5, 1e-4, 1e-3, 1e-2, 1e-1]},
        n_initial_parameters=5, max_iter=10, random_state=0)
    search.fit(from_numpy(X), from_numpy(y))
    assert search.best_score_ > 0.6
    assert search.metadata_["n_models"] == 5
    pred = search.predict(from_numpy(X)).gather().cpu().numpy()
    assert pred.shape == (2000,)


def test_successive_halving():
    X, y = _cls(n=2000, seed=6)
    sha = SuccessiveHalvingSearchCV(
        SGDClassifier(loss="log_loss", random_state=0),
        {"alpha": [1e-5, 1e-4, 1e-3, 1e-2]},
        n_initial_parameters=4, n_initial_iter=2, max_iter=12,
        random_state=0)
    sha.fit(from_numpy(X), from_numpy(y))
    assert sha.best_score_ > 0.6


def test_hyperband():
    X, y = _cls(n=2000, seed=7)
    hb = HyperbandSearchCV(
        SGDClassifier(loss="log_loss", random_state=0),
        {"alpha": [1e-5, 1e-4, 1e-3, 1e-2, 1e-1]},
        max_iter=9, random_state=0)
    meta = hb.metadata
    assert meta["n_models"] >= 5
    hb.fit(from_numpy(X), from_numpy(y))
    assert hb.best_score_ > 0.6
    assert hb.metadata_["partial_fit_calls"] > 0


def _hyperband_world2(rank, world, X_bytes, y_bytes, n, f):
    X = np.frombuffer(X_bytes, dtype=np.float32).reshape(n, f).copy()
    y = np.frombuffer(y_bytes, dtype=np.float32).copy()
    hb = HyperbandSearchCV(
        SGDClassifier(loss="log_loss", random_state=0),
        {"alpha": [1e-5, 1e-4, 1e-3, 1e-2]}, max_iter=4, random_state=0)
    hb.fit(from_numpy(X), from_numpy(y))
    assert hb.best_score_ > 0.55
    assert hb.best_params_ is not None


def test_hyperband_world2(mp_runner):
    X, y = _cls(n=1600, seed=8)
    mp_runner(_hyperband_world2, world=2,
              args=(X.tobytes(), y.tobytes(), X.shape[0], X.shape[1]))


def _grid_rank_parallel(rank, world, a_bytes, y_bytes, n, f):
    X = np.frombuffer(a_bytes, dtype=np.float32).reshape(n, f).copy()
    y = np.frombuffer(y_bytes, dtype=np.float32).copy()
    gs = GridSearchCV(LogisticRegression(solver="lbfgs", max_iter=50),
                      {"C": [0.01, 0.1, 1.0, 10.0]}, cv=3,
                      candidate_parallel="ranks")
    gs.fit(from_numpy(X), from_numpy(y))
    # same data via the all-ranks path must give identical scores
    gs2 = GridSearchCV(LogisticRegression(solver="lbfgs", max_iter=50),
                       {"C": [0.01, 0.1, 1.0, 10.0]}, cv=3,
                       candidate_parallel="none")
    gs2.fit(from_numpy(X), from_numpy(y))
    assert gs.best_params_ == gs2.best_params_
    np.testing.assert_allclose(gs.cv_results_["mean_test_score"],
                               gs2.cv_results_["mean_test_score"], atol=0.03)
    # every candidate got a real score on every fold
    assert not np.isnan(gs.cv_results_["mean_test_score"]).any()
    pred = gs.predict(from_numpy(X)).gather().cpu().numpy()
    assert pred.shape == (len(y),)


def test_grid_search_candidate_parallel_world2(mp_runner):
    X, y = _cls(1200, 5, seed=9)
    mp_runner(_grid_rank_parallel, world=2,
              args=(X.tobytes(), y.tobytes(), X.shape[0], X.shape[1]))


def test_incremental_mismatched_partitions_raises():
    from mi355ml.core.bag import DistList
    X = DistList([np.zeros((10, 3), dtype=np.float32),
                  np.zeros((5, 3), dtype=np.float32)])
    y = DistList([np.zeros(10, dtype=np.float32)])
    inc = Incremental(SGDClassifier())
    with pytest.raises(ValueError, match="partition"):
        inc.fit(X, y, classes=[0.0, 1.0])


def test_incremental_mismatched_rows_raises():
    X = from_numpy(np.zeros((20, 3), dtype=np.float32))
    y = from_numpy(np.zeros(12, dtype=np.float32))
    inc = Incremental(SGDClassifier())
    with pytest.raises(ValueError, match="shard"):
        inc.fit(X, y, classes=[0.0, 1.0])


def test_best_estimator_is_trained_winner():
    # advisor r1 (high): best_estimator_ must be the trained+scored winner,
    # not a one-pass refit. SGD's t_ counts samples seen: a refit over the
    # single train block would have t_ ~ n_train+1, the real winner (up to
    # max_iter partial_fits on the same block) far more.
    X, y = _cls(n=2000, seed=11)
    search = IncrementalSearchCV(
        SGDClassifier(loss="log_loss", random_state=0),
        {"alpha": [1e-5, 1e-4, 1e-3]},
        n_initial_parameters=3, max_iter=8, random_state=0)
    search.fit(from_numpy(X), from_numpy(y))
    n_train = int(2000 * 0.85)
    assert search.best_estimator_.t_ - 1 >= 2 * n_train
    # and the winner's recorded budget matches a multi-round survivor
    pf = search.cv_results_["partial_fit_calls"]
    assert pf[search.best_index_] == pf.max()


def test_sha_rung_budget_is_target_total():
    # advisor r1 (medium): rung budget r0*eta^rung is the TOTAL partial_fit
    # target, so with r0=2, eta=3, max_iter=18 the rung totals are exactly
    # {2, 6, 18} and nobody trains past 18.
    X, y = _cls(n=1200, seed=12)
    sha = SuccessiveHalvingSearchCV(
        SGDClassifier(loss="log_loss", random_state=0),
        {"alpha": [1e-5, 3e-5, 1e-4, 3e-4, 1e-3, 3e-3, 1e-2, 3e-2, 1e-1]},
        n_initial_parameters=9, n_initial_iter=2, max_iter=18,
        random_state=0)
    sha.fit(from_numpy(X), from_numpy(y))
    pf = sha.cv_results_["partial_fit_calls"]
    assert set(pf.tolist()) <= {2, 6, 18}
    assert pf.max() == 18


class _BonusDummy:
    """score = bonus * mean(y == 1): deterministic, data-dependent, no
    learning — isolates WHERE scoring happens (local vs global val set)."""

    def __init__(self, bonus=0.0):
        self.bonus = bonus

    def get_params(self, deep=True):
        return {"bonus": self.bonus}

    def set_params(self, **p):
        for k, v in p.items():
            setattr(self, k, v)
        return self

    def partial_fit(self, X, y, classes=None):
        return self

    def score(self, X, y):
        b = self.bonus[0] if isinstance(self.bonus, list) else self.bonus
        return float((np.asarray(y) == 1).mean()) * b


def _skewed_scoring_world2(rank, world, n):
    # rank 0's shard is (almost) all y=1, rank 1's all y=0. Owner-local
    # scoring would hand trial 0 (bonus=1, owner rank 0) a score ~1.0 and
    # trial 1 (bonus=2, owner rank 1) ~0.0 — the wrong winner. Global
    # count-weighted scoring gives both trials the same mean(y==1)~0.5
    # factor, so bonus=2 must win.
    rng = np.random.default_rng(0)
    X = rng.standard_normal((n, 4)).astype(np.float32)
    y = np.concatenate([np.ones(n // 2, dtype=np.float32),
                        np.zeros(n - n // 2, dtype=np.float32)])
    search = IncrementalSearchCV(
        _BonusDummy(), {"bonus": [1.0, 2.0]},
        n_initial_parameters="grid", max_iter=2, random_state=0)
    search.fit(from_numpy(X), from_numpy(y))
    assert search.best_params_ == {"bonus": 2.0}
    assert abs(search.best_score_ - 2.0 * 0.5) < 0.15


def test_incremental_global_scoring_skewed_shards_world2(mp_runner):
    mp_runner(_skewed_scoring_world2, world=2, args=(1000,))


def test_grid_search_return_train_score():
    X, y = _cls(n=1500, seed=13)
    gs = GridSearchCV(LogisticRegression(solver="lbfgs", max_iter=50),
                      {"C": [0.1, 1.0]}, cv=3, return_train_score=True)
    gs.fit(from_numpy(X), from_numpy(y))
    assert "mean_train_score" in gs.cv_results_
    assert "split0_train_score" in gs.cv_results_
    tr = gs.cv_results_["mean_train_score"]
    te = gs.cv_results_["mean_test_score"]
    assert tr.shape == te.shape == (2,)
    assert not np.isnan(tr).any()
    # train score should be >= test score on average (same-data fit)
    assert tr.mean() >= te.mean() - 0.05


class _FailingEstimator:
    """Fails to fit whenever C is negative (error_score machinery)."""

    def __init__(self, C=1.0):
        self.C = C

    def get_params(self, deep=True):
        return {"C": self.C}

    def set_params(self, **p):
        for k, v in p.items():
            setattr(self, k, v)
        return self

    def fit(self, X, y=None):
        if self.C < 0:
            raise ValueError("boom")
        self.mean_ = 0.0
        return self

    def score(self, X, y=None):
        return 0.5 + 0.1 * self.C


def test_grid_search_error_score():
    X, y = _cls(n=400, f=4, seed=14)
    gs = GridSearchCV(_FailingEstimator(), {"C": [-1.0, 1.0, 2.0]}, cv=2,
                      error_score=-999.0, candidate_parallel="none")
    with pytest.warns(UserWarning, match="failed on fold"):
        gs.fit(from_numpy(X), from_numpy(y))
    assert gs.cv_results_["mean_test_score"][0] == -999.0
    assert gs.best_params_ == {"C": 2.0}
    # failed candidate ranked last
    assert gs.cv_results_["rank_test_score"][0] == 3
    # error_score='raise' re-raises
    gs2 = GridSearchCV(_FailingEstimator(), {"C": [-1.0, 1.0]}, cv=2,
                       error_score="raise", candidate_parallel="none")
    with pytest.raises(ValueError, match="boom"):
        gs2.fit(from_numpy(X), from_numpy(y))


def test_grid_search_multimetric():
    X, y = _cls(n=1500, seed=15)
    gs = GridSearchCV(LogisticRegression(solver="lbfgs", max_iter=50),
                      {"C": [0.1, 1.0]},
                      scoring={"acc": "accuracy", "nll": "neg_log_loss"},
                      cv=3, refit="acc", return_train_score=True)
    gs.fit(from_numpy(X), from_numpy(y))
    for key in ("mean_test_acc", "mean_test_nll", "rank_test_acc",
                "split0_test_nll", "mean_train_acc"):
        assert key in gs.cv_results_, key
    assert gs.multimetric_
    assert gs.best_score_ == gs.cv_results_["mean_test_acc"][gs.best_index_]
    assert hasattr(gs, "best_estimator_")
    # refit must name a metric under multimetric scoring
    bad = GridSearchCV(LogisticRegression(), {"C": [1.0]},
                       scoring=["accuracy", "neg_log_loss"], cv=2)
    with pytest.raises(ValueError, match="refit"):
        bad.fit(from_numpy(X), from_numpy(y))


def _stealing_world2(rank, world, n):
    # 8 candidates; the two best (tids 0 and 2, highest bonus) are both
    # rank-0-pinned under tid%world. The rebalance must spread the round-2
    # survivors across BOTH ranks (transfer is free: models already
    # traveled for global scoring), and best_estimator_ must still be the
    # trained winner broadcast from its CURRENT owner.
    rng = np.random.default_rng(1)
    X = rng.standard_normal((n, 4)).astype(np.float32)
    y = (rng.uniform(size=n) < 0.5).astype(np.float32)
    bonuses = [8.0, 1.0, 7.0, 2.0, 3.0, 4.0, 5.0, 6.0]
    sha = SuccessiveHalvingSearchCV(
        _BonusDummy(), {"bonus": [[b] for b in bonuses]},
        n_initial_parameters="grid", n_initial_iter=1, max_iter=9,
        random_state=0)
    # grid over a list-valued param: each candidate gets one bonus value
    sha.fit(from_numpy(X), from_numpy(y))
    assert sha.best_params_["bonus"] == [8.0]
    survivors = [tid for tid, b in enumerate(bonuses) if b >= 7.0]
    owners = {tid: sha._owner_of[tid] for tid in survivors}
    assert set(owners.values()) == {0, 1}, owners
    assert isinstance(sha.best_estimator_, _BonusDummy)
    assert sha.best_estimator_.bonus == [8.0]


def test_trial_stealing_rebalances_world2(mp_runner):
    mp_runner(_stealing_world2, world=2, args=(600,))


def test_search_threads_identical_to_serial():
    # trial fan-out over threads must be bit-identical to serial (no
    # shared state between trials)
    import mi355ml.config as config
    X, y = _cls(n=1500, seed=16)
    def run():
        s = IncrementalSearchCV(
            SGDClassifier(loss="log_loss", random_state=0),
            {"alpha": [1e-5, 1e-4, 1e-3, 1e-2]},
            n_initial_parameters=4, max_iter=6, random_state=0)
        s.fit(from_numpy(X), from_numpy(y))
        return s
    with config.set({"search.threads": 0}):
        a = run()
    with config.set({"search.threads": 8}):
        b = run()
    assert a.best_params_ == b.best_params_
    np.testing.assert_array_equal(a.cv_results_["test_score"],
                                  b.cv_results_["test_score"])
    np.testing.assert_array_equal(a.best_estimator_.coef_,
                                  b.best_estimator_.coef_)


def test_balance_trials_properties():
    from mi355ml.model_selection._incremental import _balance_trials
    budgets = {0: 9, 1: 3, 2: 3, 3: 1, 4: 1, 5: 1}
    a = _balance_trials(budgets, 2)
    loads = [sum(budgets[t] for t, r in a.items() if r == i)
             for i in range(2)]
    assert abs(loads[0] - loads[1]) <= max(budgets.values())
    assert a == _balance_trials(budgets, 2)  # deterministic
    # more ranks than trials: everyone gets at most one
    b = _balance_trials({0: 2, 1: 2}, 8)
    assert sorted(b.values()) == [0, 1]


def test_hyperband_world4(mp_runner):
    # stealing + global scoring + threaded trials under 4 ranks
    X, y = _cls(n=1600, seed=18)
    mp_runner(_hyperband_world2, world=4,
              args=(X.tobytes(), y.tobytes(), X.shape[0], X.shape[1]))
