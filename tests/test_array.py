"""DistArray reductions vs numpy oracle, single- and multi-process."""

import numpy as np
import pytest
import torch

import mi355ml
from mi355ml.core.array import from_numpy, shard_rows


def _data(n=1000, f=7, seed=0):
    rng = np.random.default_rng(seed)
    return (rng.standard_normal((n, f)) * rng.uniform(0.5, 3, f)
            + rng.uniform(-2, 2, f)).astype(np.float32)


def test_moments_single():
    a = _data()
    X = from_numpy(a)
    np.testing.assert_allclose(X.mean().cpu().numpy(), a.mean(0), rtol=1e-6,
                               atol=1e-6)
    np.testing.assert_allclose(X.var().cpu().numpy(), a.var(0), rtol=1e-5,
                               atol=1e-6)
    np.testing.assert_allclose(X.var(ddof=1).cpu().numpy(), a.var(0, ddof=1),
                               rtol=1e-5, atol=1e-6)
    np.testing.assert_allclose(X.sum().cpu().numpy(), a.sum(0, dtype=np.float64),
                               rtol=1e-6)


def test_minmax_gather():
    a = _data()
    X = from_numpy(a)
    np.testing.assert_allclose(X.min().cpu().numpy(), a.min(0))
    np.testing.assert_allclose(X.max().cpu().numpy(), a.max(0))
    np.testing.assert_allclose(X.gather().cpu().numpy(), a)


def test_dot_xtdot():
    a = _data()
    X = from_numpy(a)
    w = np.random.default_rng(1).standard_normal(a.shape[1]).astype(np.float32)
    z = X.dot(torch.from_numpy(w))
    np.testing.assert_allclose(z.gather().cpu().numpy(), a @ w, rtol=1e-4,
                               atol=1e-4)
    v = from_numpy((a @ w).astype(np.float32))
    g = X.xt_dot(v)
    np.testing.assert_allclose(g.cpu().numpy()[:, 0],
                               a.astype(np.float64).T @ (a @ w).astype(np.float64),
                               rtol=1e-5)


def test_1d_moments():
    rng = np.random.default_rng(2)
    a = rng.standard_normal(500).astype(np.float32)
    y = from_numpy(a)
    assert y.shape == (500,)
    np.testing.assert_allclose(float(y.mean()), a.mean(), rtol=1e-6, atol=1e-6)
    np.testing.assert_allclose(float(y.var()), a.var(), rtol=1e-5, atol=1e-6)


def _dist_moments_body(rank, world, a_bytes, shape):
    a = np.frombuffer(a_bytes, dtype=np.float32).reshape(shape).copy()
    X = from_numpy(a)
    assert X.local.shape[0] < a.shape[0]
    np.testing.assert_allclose(X.mean().cpu().numpy(), a.mean(0), rtol=1e-6,
                               atol=1e-6)
    np.testing.assert_allclose(X.var().cpu().numpy(), a.var(0), rtol=1e-5,
                               atol=1e-6)
    np.testing.assert_allclose(X.min().cpu().numpy(), a.min(0))
    np.testing.assert_allclose(X.gather().cpu().numpy(), a)


def test_dist_moments_world2(mp_runner):
    a = _data(501, 5, seed=3)  # uneven split
    mp_runner(_dist_moments_body, world=2, args=(a.tobytes(), a.shape))


def _rebalance_body(rank, world, a_bytes, shape):
    import torch
    from mi355ml.core.array import DistArray, from_numpy
    a = np.frombuffer(a_bytes, dtype=np.float32).reshape(shape).copy()
    X = from_numpy(a)
    # make shards uneven: rank 0 keeps 10 rows, rank 1 the rest
    keep = 10 if rank == 0 else X.local.shape[0]
    from mi355ml.core.array import from_local
    un = from_local(X.local[:keep].contiguous())
    rb = un.rebalance()
    w = un.comm.world_size
    assert abs(rb.local.shape[0] - un.global_rows / w) <= 1
    # content preserved in global order
    expect = np.concatenate([a[:10], a[shape[0] // 2:]])
    np.testing.assert_array_equal(rb.gather().cpu().numpy(), expect)


def test_rebalance_world2(mp_runner):
    a = _data(400, 3, seed=9)
    mp_runner(_rebalance_body, world=2, args=(a.tobytes(), a.shape))


def test_histogram_topk_getitem():
    rng = np.random.default_rng(11)
    a = rng.standard_normal(5000).astype(np.float32)
    y = from_numpy(a)
    h, edges = y.histogram(bins=20, range=(-4, 4))
    ref, redges = np.histogram(a, bins=20, range=(-4, 4))
    np.testing.assert_array_equal(h.cpu().numpy(), ref)
    np.testing.assert_allclose(edges.cpu().numpy(), redges, rtol=1e-6,
                               atol=1e-12)
    tk = y.topk(7).cpu().numpy()
    np.testing.assert_allclose(tk, np.sort(a)[-7:][::-1], rtol=1e-6)
    tk = y.topk(3, largest=False).cpu().numpy()
    np.testing.assert_allclose(tk, np.sort(a)[:3], rtol=1e-6)
    M = rng.standard_normal((100, 4)).astype(np.float32)
    X = from_numpy(M)
    sl = X[10:37]
    assert sl.shape == (27, 4)
    np.testing.assert_array_equal(sl.gather().cpu().numpy(), M[10:37])


def _hist_world2(rank, world, a_bytes, n):
    a = np.frombuffer(a_bytes, dtype=np.float32).copy()
    y = from_numpy(a)
    h, _ = y.histogram(bins=10, range=(-3, 3))
    ref, _ = np.histogram(a, bins=10, range=(-3, 3))
    np.testing.assert_array_equal(h.cpu().numpy(), ref)
    sl = y[100:301]
    np.testing.assert_array_equal(sl.gather().cpu().numpy(), a[100:301])


def test_hist_world2(mp_runner):
    rng = np.random.default_rng(12)
    a = rng.standard_normal(1001).astype(np.float32)
    mp_runner(_hist_world2, world=2, args=(a.tobytes(), len(a)))


def _tiny_world2(rank, world):
    import torch
    from mi355ml.preprocessing import StandardScaler
    from mi355ml.cluster import KMeans
    from mi355ml.linear_model import LogisticRegression
    # 3 rows over 2 ranks: rank 1 gets 1 row; 1-row arrays everywhere
    a = np.array([[1.0, 2.0], [3.0, 4.0], [5.0, 6.0]], dtype=np.float32)
    y = np.array([0.0, 1.0, 1.0], dtype=np.float32)
    X = from_numpy(a)
    ss = StandardScaler().fit(X)
    np.testing.assert_allclose(ss.mean_.cpu().numpy(), a.mean(0), rtol=1e-6)
    lr = LogisticRegression(C=1.0, max_iter=20).fit(X, from_numpy(y))
    assert lr.coef_.shape == (2,)
    km = KMeans(n_clusters=2, random_state=0, max_iter=5).fit(X)
    assert km.cluster_centers_.shape == (2, 2)


def test_tiny_data_world2(mp_runner):
    mp_runner(_tiny_world2, world=2)


def _empty_rank_world2(rank, world):
    # fewer rows than ranks would give rank 1 zero rows at world 3; at
    # world 2 use a 1-row array so rank 1 is empty
    a = np.array([[7.0, 8.0, 9.0]], dtype=np.float32)
    X = from_numpy(a)
    if rank == 1:
        assert X.local.shape[0] == 0
    np.testing.assert_allclose(X.mean().cpu().numpy(), a.mean(0), rtol=1e-6)
    np.testing.assert_allclose(X.min().cpu().numpy(), a.min(0))
    np.testing.assert_array_equal(X.gather().cpu().numpy(), a)


def test_empty_rank_world2(mp_runner):
    mp_runner(_empty_rank_world2, world=2)


def _take_concat_cum_world2(rank, world, n, f):
    import torch
    from mi355ml.core.array import concat, from_numpy, take
    rng = np.random.default_rng(17)
    a = rng.standard_normal((n, f)).astype(np.float32)
    b = rng.standard_normal((n // 2, f)).astype(np.float32)
    A, B = from_numpy(a), from_numpy(b)
    # fancy take with duplicates, negatives, cross-shard pulls
    idx = rng.integers(-n, n, 3 * n // 2)
    got = take(A, idx).gather().cpu().numpy()
    np.testing.assert_array_equal(got, a[idx])
    # result is canonically sharded
    T = take(A, idx)
    assert T.local.shape[0] in ((len(idx)) // world, len(idx) // world + 1)
    # concat
    C = concat([A, B])
    np.testing.assert_array_equal(C.gather().cpu().numpy(),
                                  np.vstack([a, b]))
    assert C.global_rows == n + n // 2
    # cumsum / cumprod across the shard boundary
    np.testing.assert_allclose(A.cumsum().gather().cpu().numpy(),
                               a.cumsum(axis=0), rtol=1e-5, atol=1e-4)
    p = np.abs(a[:40]) + 0.5
    P = from_numpy(p)
    np.testing.assert_allclose(P.cumprod().gather().cpu().numpy(),
                               p.cumprod(axis=0), rtol=1e-4, atol=1e-4)


def test_take_concat_cum_world2(mp_runner):
    mp_runner(_take_concat_cum_world2, world=2, args=(120, 3))


def test_take_out_of_range_raises():
    from mi355ml.core.array import from_numpy, take
    A = from_numpy(np.zeros((10, 2), dtype=np.float32))
    with pytest.raises(IndexError):
        take(A, [10])
