"""Depth tests: ops-level tsqr, ADMM across ranks, distributed vocabulary,
scorer registry, adaptive-search patience, config layering."""

import numpy as np
import pytest
import torch

import mi355ml
from mi355ml.core.array import from_numpy
from mi355ml.core.bag import DistList


def _lowrank(n=1200, d=12, r=4, seed=0):
    rng = np.random.default_rng(seed)
    return (rng.standard_normal((n, r)) @ rng.standard_normal((r, d))
            + 0.01 * rng.standard_normal((n, d))).astype(np.float32)


def test_tsqr_ops_level():
    from mi355ml.ops.linalg import tsqr
    a = _lowrank()
    X = from_numpy(a)
    Q, R = tsqr(X)
    Qm = Q.gather().cpu().numpy()
    Rm = R.cpu().numpy()
    np.testing.assert_allclose(Qm @ Rm, a, rtol=1e-3, atol=1e-3)
    np.testing.assert_allclose(Qm.T @ Qm, np.eye(a.shape[1]), atol=1e-4)
    # R-only path: same factorization up to fp (compare RtR = XtX, the
    # invariant — R's tail entries are noise-scale on this low-rank matrix)
    Qn, R2 = tsqr(X, compute_q=False)
    assert Qn is None
    R2m = R2.cpu().numpy()
    np.testing.assert_allclose(R2m.T @ R2m, a.T @ a, rtol=1e-3,
                               atol=1e-2 * np.abs(a.T @ a).max())


def test_tsqr_blocked_stage1():
    """Force multiple stage-1 blocks (the A.1 recursion path)."""
    from mi355ml.ops import linalg as L
    orig = L._block_rows_for
    L._block_rows_for = lambda d: 100
    try:
        a = _lowrank(n=950, d=8, seed=1)
        Q, R = L.tsqr(from_numpy(a))
        Qm = Q.gather().cpu().numpy()
        np.testing.assert_allclose(Qm @ R.cpu().numpy(), a, rtol=1e-3,
                                   atol=1e-3)
        np.testing.assert_allclose(Qm.T @ Qm, np.eye(8), atol=1e-4)
    finally:
        L._block_rows_for = orig


def _admm_world2(rank, world, X_bytes, y_bytes, n, f):
    from mi355ml.linear_model import LogisticRegression
    X = np.frombuffer(X_bytes, dtype=np.float32).reshape(n, f).copy()
    y = np.frombuffer(y_bytes, dtype=np.float32).copy()
    lr = LogisticRegression(C=100.0, solver="admm", max_iter=25).fit(
        from_numpy(X), from_numpy(y))
    acc = lr.score(from_numpy(X), from_numpy(y))
    assert acc > 0.8, acc


def test_admm_world2(mp_runner):
    rng = np.random.default_rng(2)
    X = rng.standard_normal((1501, 5)).astype(np.float32)
    w = rng.standard_normal(5) * 2
    y = ((1 / (1 + np.exp(-(X @ w)))) > rng.uniform(size=1501)).astype(
        np.float32)
    mp_runner(_admm_world2, world=2,
              args=(X.tobytes(), y.tobytes(), X.shape[0], X.shape[1]))


def _countvec_world2(rank, world):
    from mi355ml.feature_extraction.text import CountVectorizer
    docs = ["alpha beta", "beta gamma", "gamma delta", "delta alpha"]
    dl = DistList.from_global(docs)
    assert len(dl.local) == 2  # actually sharded
    cv = CountVectorizer().fit(dl)
    assert sorted(cv.vocabulary_) == ["alpha", "beta", "delta", "gamma"]
    out = cv.transform(dl)
    import scipy.sparse as sp
    m = sp.vstack(out.gather())
    assert m.shape == (4, 4)
    assert m.sum() == 8


def test_countvec_world2(mp_runner):
    mp_runner(_countvec_world2, world=2)


def test_scorer_registry():
    from mi355ml.metrics import SCORERS, get_scorer, check_scoring
    from mi355ml.linear_model import LinearRegression
    rng = np.random.default_rng(3)
    X = rng.standard_normal((500, 3)).astype(np.float32)
    y = (X @ np.array([1.0, -2.0, 0.5]) + 0.01
         * rng.standard_normal(500)).astype(np.float32)
    lr = LinearRegression(C=1e8, max_iter=200).fit(from_numpy(X),
                                                   from_numpy(y))
    r2 = get_scorer("r2")(lr, from_numpy(X), from_numpy(y))
    assert r2 > 0.999
    nmse = SCORERS["neg_mean_squared_error"](lr, from_numpy(X),
                                             from_numpy(y))
    assert -1e-2 < nmse <= 0
    with pytest.raises(ValueError):
        get_scorer("nope")
    default = check_scoring(lr, None)
    assert abs(default(lr, from_numpy(X), from_numpy(y)) - r2) < 1e-9


def test_incremental_search_patience():
    from sklearn.linear_model import SGDClassifier
    from mi355ml.model_selection import IncrementalSearchCV
    rng = np.random.default_rng(4)
    X = rng.standard_normal((1500, 5)).astype(np.float32)
    w = rng.standard_normal(5)
    y = ((X @ w) > 0).astype(np.float32)
    s = IncrementalSearchCV(
        SGDClassifier(loss="log_loss", random_state=0),
        {"alpha": [1e-4, 1e-3]}, n_initial_parameters=2, decay_rate=0,
        patience=2, tol=0.0, max_iter=50, random_state=0)
    s.fit(from_numpy(X), from_numpy(y))
    # patience must stop well before max_iter rounds
    assert s.n_iter_ < 50
    assert s.best_score_ > 0.7


def test_config_layering(monkeypatch):
    monkeypatch.setenv("MI355ML_ARRAY__BLOCK_ROWS", "12345")
    mi355ml.config.refresh()
    try:
        assert mi355ml.config.get("array.block_rows") == 12345
        with mi355ml.config.set({"array.block_rows": 99}):
            assert mi355ml.config.get("array.block_rows") == 99
        assert mi355ml.config.get("array.block_rows") == 12345
    finally:
        monkeypatch.delenv("MI355ML_ARRAY__BLOCK_ROWS")
        mi355ml.config.refresh()
    assert mi355ml.config.get("comm.backend") == "auto"
    with pytest.raises(KeyError):
        mi355ml.config.get("no.such.key")
    assert mi355ml.config.get("no.such.key", 7) == 7


def test_column_transformer():
    from mi355ml.compose import ColumnTransformer, make_column_transformer
    from mi355ml.preprocessing import MinMaxScaler, StandardScaler
    rng = np.random.default_rng(5)
    X = (rng.standard_normal((400, 5)) * [1, 10, 100, 5, 2]).astype(
        np.float32)
    ct = ColumnTransformer(
        [("a", StandardScaler(), [0, 1]), ("b", MinMaxScaler(), [2])],
        remainder="passthrough")
    out = ct.fit_transform(from_numpy(X)).gather().cpu().numpy()
    assert out.shape == (400, 5)
    np.testing.assert_allclose(out[:, :2].mean(axis=0), 0, atol=1e-4)
    assert out[:, 2].min() >= -1e-6 and out[:, 2].max() <= 1 + 1e-6
    np.testing.assert_allclose(out[:, 3:], X[:, 3:], rtol=1e-6)
    ct2 = make_column_transformer((StandardScaler(), [0]), remainder="drop")
    out2 = ct2.fit_transform(from_numpy(X)).gather().cpu().numpy()
    assert out2.shape == (400, 1)
