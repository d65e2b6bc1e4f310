"""Wide-X feature-dim sharding (VERDICT r1 #5 / SURVEY §5.7): ColDistArray
ops, col-sharded GLM (lbfgs et al.) and TruncatedSVD vs the row-sharded
oracle, on CPU (world 1) and gloo world 2."""

import numpy as np
import pytest
import torch

from mi355ml.core.array import from_numpy
from mi355ml.core.colarray import (ColDistArray, Replicated, from_numpy_cols,
                                   shard_cols)
from mi355ml.decomposition import TruncatedSVD
from mi355ml.linear_model import LinearRegression, LogisticRegression


def _data(n=800, d=64, seed=0):
    rng = np.random.default_rng(seed)
    X = rng.standard_normal((n, d)).astype(np.float32)
    w = rng.standard_normal(d).astype(np.float32) * 0.5
    p = 1 / (1 + np.exp(-(X @ w)))
    y = (rng.uniform(size=n) < p).astype(np.float32)
    return X, y


def test_colarray_ops_world1():
    X, _ = _data()
    A = from_numpy_cols(X)
    assert A.shape == X.shape
    w = np.random.default_rng(1).standard_normal(X.shape[1]).astype(
        np.float32)
    np.testing.assert_allclose(A.dot(torch.from_numpy(w)).numpy(),
                               X @ w, rtol=1e-4, atol=1e-4)
    r = np.random.default_rng(2).standard_normal(X.shape[0]).astype(
        np.float32)
    np.testing.assert_allclose(A.rmatvec(torch.from_numpy(r)).numpy(),
                               X.T @ r, rtol=1e-3, atol=1e-3)
    np.testing.assert_allclose(A.mean(axis=0).numpy(), X.mean(axis=0),
                               rtol=1e-4, atol=1e-5)
    np.testing.assert_allclose(A.gather().numpy(), X)


def test_col_logreg_matches_row_sharded_world1():
    X, y = _data()
    m_row = LogisticRegression(solver="lbfgs", max_iter=60).fit(
        from_numpy(X), from_numpy(y))
    m_col = LogisticRegression(solver="lbfgs", max_iter=60).fit(
        from_numpy_cols(X), torch.from_numpy(y))
    np.testing.assert_allclose(m_col.coef_.numpy(), m_row.coef_.numpy(),
                               rtol=5e-3, atol=5e-3)
    # predict through the Replicated path
    pred = m_col.predict(from_numpy_cols(X)).gather().numpy()
    pred_row = m_row.predict(from_numpy(X)).gather().numpy()
    assert (pred == pred_row).mean() > 0.99
    acc = m_col.score(from_numpy_cols(X), y)
    assert acc > 0.75


def test_col_solvers_unsupported_raise():
    X, y = _data(n=200, d=16)
    for solver, msg in (("newton", "wide"), ("admm", "row shards")):
        with pytest.raises(ValueError, match=msg):
            LogisticRegression(solver=solver, max_iter=5).fit(
                from_numpy_cols(X), torch.from_numpy(y))


def test_col_tsvd_matches_dense_svd_world1():
    rng = np.random.default_rng(3)
    # low-rank + noise so the spectrum is recoverable
    U = rng.standard_normal((600, 10)).astype(np.float32)
    V = rng.standard_normal((10, 200)).astype(np.float32)
    X = U @ V + 0.01 * rng.standard_normal((600, 200)).astype(np.float32)
    k = 8
    t = TruncatedSVD(n_components=k, algorithm="randomized", n_iter=4,
                     random_state=0)
    Z = t.fit_transform(from_numpy_cols(X))
    assert isinstance(Z, Replicated)
    s_exact = np.linalg.svd(X, compute_uv=False)[:k]
    np.testing.assert_allclose(t.singular_values_.numpy(), s_exact,
                               rtol=2e-2)
    # subspace agreement with exact right singular vectors
    _, _, Vh = np.linalg.svd(X, full_matrices=False)
    M = t.components_.numpy() @ Vh[:k].T
    np.testing.assert_allclose(np.abs(np.diag(M)), np.ones(k), atol=5e-2)
    # transform equals X @ components_.T
    np.testing.assert_allclose(
        Z.gather().numpy(), X @ t.components_.numpy().T,
        rtol=1e-3, atol=1e-2)


def _col_vs_row_world2(rank, world, X_bytes, y_bytes, n, f):
    X = np.frombuffer(X_bytes, dtype=np.float32).reshape(n, f).copy()
    y = np.frombuffer(y_bytes, dtype=np.float32).copy()
    # col-sharded fit (every rank passes full X; keeps its column slice)
    m_col = LogisticRegression(solver="lbfgs", max_iter=60).fit(
        from_numpy_cols(X), torch.from_numpy(y))
    # row-sharded oracle on the same global data
    m_row = LogisticRegression(solver="lbfgs", max_iter=60).fit(
        from_numpy(X), from_numpy(y))
    np.testing.assert_allclose(m_col.coef_.numpy(), m_row.coef_.numpy(),
                               rtol=1e-2, atol=1e-2)
    # TSVD: col-sharded vs single-process oracle
    t_col = TruncatedSVD(n_components=6, algorithm="randomized", n_iter=4,
                         random_state=0)
    t_col.fit(from_numpy_cols(X))
    s_exact = np.linalg.svd(X, compute_uv=False)[:6]
    np.testing.assert_allclose(t_col.singular_values_.numpy(), s_exact,
                               rtol=5e-2)
    # component vectors live full-length on every rank and are identical
    assert t_col.components_.shape == (6, f)


def test_col_vs_row_world2(mp_runner):
    X, y = _data(n=600, d=48, seed=4)
    mp_runner(_col_vs_row_world2, world=2,
              args=(X.tobytes(), y.tobytes(), X.shape[0], X.shape[1]))


def _col_shard_layout_world2(rank, world, n, f):
    rng = np.random.default_rng(5)
    X = rng.standard_normal((n, f)).astype(np.float32)
    A = from_numpy_cols(X)
    # shard bounds partition the feature axis
    assert A.global_cols == f
    assert A.local_cols in (f // world, f // world + 1)
    np.testing.assert_allclose(A.gather().numpy(), X, rtol=0, atol=0)
    w = rng.standard_normal(f).astype(np.float32)
    np.testing.assert_allclose(A.dot(torch.from_numpy(w)).cpu().numpy(),
                               X @ w, rtol=1e-3, atol=1e-3)
    r = rng.standard_normal(n).astype(np.float32)
    np.testing.assert_allclose(A.rmatvec(torch.from_numpy(r)).cpu().numpy(),
                               X.T @ r, rtol=1e-3, atol=1e-2)


def test_col_shard_layout_world2(mp_runner):
    mp_runner(_col_shard_layout_world2, world=2, args=(300, 33))


def test_col_pca_matches_dense_world1():
    from mi355ml.decomposition import PCA
    rng = np.random.default_rng(7)
    U = rng.standard_normal((500, 6)).astype(np.float32) * 3
    V = rng.standard_normal((6, 120)).astype(np.float32)
    X = U @ V + 0.05 * rng.standard_normal((500, 120)).astype(np.float32) \
        + rng.standard_normal(120).astype(np.float32)[None, :]
    k = 5
    p = PCA(n_components=k, svd_solver="randomized", iterated_power=4,
            random_state=0)
    Z = p.fit_transform(from_numpy_cols(X))
    from sklearn.decomposition import PCA as SkPCA
    sk = SkPCA(n_components=k, svd_solver="full").fit(X.astype(np.float64))
    np.testing.assert_allclose(p.singular_values_.numpy(),
                               sk.singular_values_, rtol=2e-2)
    np.testing.assert_allclose(p.mean_.numpy(), X.mean(axis=0), atol=1e-3)
    # subspace agreement
    M = p.components_.numpy() @ sk.components_.T
    np.testing.assert_allclose(np.abs(np.diag(M)), np.ones(k), atol=5e-2)
    # transform path (centering + projection + allreduce)
    Zt = p.transform(from_numpy_cols(X)).gather().numpy()
    ref = (X - X.mean(axis=0)) @ p.components_.numpy().T
    np.testing.assert_allclose(Zt, ref, rtol=1e-3, atol=1e-2)
    np.testing.assert_allclose(Z.gather().numpy(), Zt, rtol=1e-3, atol=5e-2)


def _col_pca_world2(rank, world, X_bytes, n, f):
    from mi355ml.decomposition import PCA
    X = np.frombuffer(X_bytes, dtype=np.float32).reshape(n, f).copy()
    p = PCA(n_components=4, svd_solver="randomized", iterated_power=4,
            random_state=0)
    p.fit(from_numpy_cols(X))
    s_exact = np.linalg.svd(X - X.mean(axis=0),
                            compute_uv=False)[:4]
    np.testing.assert_allclose(p.singular_values_.numpy(), s_exact,
                               rtol=5e-2)
    assert p.components_.shape == (4, f)


def test_col_pca_world2(mp_runner):
    rng = np.random.default_rng(8)
    U = rng.standard_normal((400, 4)).astype(np.float32) * 2
    V = rng.standard_normal((4, 60)).astype(np.float32)
    X = (U @ V + 0.05 * rng.standard_normal((400, 60))).astype(np.float32)
    mp_runner(_col_pca_world2, world=2, args=(X.tobytes(), 400, 60))


def test_row_sharded_estimators_reject_col_input():
    """Row-shard-only estimators must refuse ColDistArray loudly — a
    column slice silently treated as a row shard would be WRONG at
    world > 1."""
    from mi355ml.cluster import KMeans
    from mi355ml.model_selection import GridSearchCV, train_test_split
    from mi355ml.preprocessing import StandardScaler
    X, y = _data(n=100, d=8)
    C = from_numpy_cols(X)
    with pytest.raises(TypeError, match="column-sharded"):
        KMeans(n_clusters=2, max_iter=2).fit(C)
    with pytest.raises(TypeError, match="column-sharded"):
        StandardScaler().fit(C)
    with pytest.raises(TypeError, match="column-sharded"):
        train_test_split(C, torch.from_numpy(y))
    with pytest.raises(TypeError, match="column-sharded"):
        GridSearchCV(LogisticRegression(), {"C": [1.0]}, cv=2).fit(
            C, torch.from_numpy(y))


def test_col_partial_fit_converges():
    X, y = _data(n=600, d=32, seed=9)
    m = LogisticRegression(C=100.0, eta0=2.0)
    C = from_numpy_cols(X)
    yt = torch.from_numpy(y)
    for _ in range(60):
        m.partial_fit(C, yt)
    assert m.score(C, y) > 0.8
