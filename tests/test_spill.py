"""Host-spill tier: streamed reductions and estimator fits match resident."""

import numpy as np
import torch

from mi355ml.core.array import from_numpy
from mi355ml.core.spill import spill_rows, to_spilled
from mi355ml.cluster import KMeans
from mi355ml.linear_model import LogisticRegression


def _data(n=4000, f=6, seed=0):
    rng = np.random.default_rng(seed)
    return rng.standard_normal((n, f)).astype(np.float32)


def test_spilled_moments_match():
    a = _data()
    X = from_numpy(a)
    S = spill_rows(torch.from_numpy(a), block_rows=700)
    n1, m1, v1 = X.moments()
    n2, m2, v2 = S.moments()
    assert n1 == n2
    torch.testing.assert_close(m1, m2, rtol=1e-10, atol=1e-8)
    torch.testing.assert_close(v1, v2, rtol=1e-8, atol=1e-6)
    torch.testing.assert_close(X.min(), S.min())
    torch.testing.assert_close(X.max(), S.max())


def test_spilled_logreg_matches_resident():
    rng = np.random.default_rng(1)
    a = _data(3000, 5, seed=1)
    w = rng.standard_normal(5)
    y = ((1 / (1 + np.exp(-(a @ w)))) > rng.uniform(size=3000)).astype(
        np.float32)
    lr1 = LogisticRegression(C=1.0, max_iter=100).fit(from_numpy(a),
                                                      from_numpy(y))
    S = spill_rows(torch.from_numpy(a), block_rows=800)
    Sy = from_numpy(y)
    lr2 = LogisticRegression(C=1.0, max_iter=100).fit(S, Sy)
    np.testing.assert_allclose(lr1.coef_.cpu().numpy(), lr2.coef_.cpu().numpy(),
                               rtol=1e-4, atol=1e-5)


def test_spilled_kmeans():
    rng = np.random.default_rng(2)
    centers = rng.uniform(-8, 8, (4, 5))
    lab = rng.integers(0, 4, 3000)
    a = (centers[lab] + rng.standard_normal((3000, 5))).astype(np.float32)
    km1 = KMeans(n_clusters=4, random_state=0).fit(from_numpy(a))
    S = spill_rows(torch.from_numpy(a), block_rows=640)
    km2 = KMeans(n_clusters=4, random_state=0).fit(S)
    # the spilled init draws candidates per block, so cluster INDEX order
    # can differ: compare as center SETS + converged inertia
    c1 = km1.cluster_centers_.cpu().numpy()
    c2 = km2.cluster_centers_.cpu().numpy()
    d = np.linalg.norm(c1[:, None, :] - c2[None, :, :], axis=2)
    assert d.min(axis=1).max() < 0.3, d.min(axis=1)
    assert abs(km1.inertia_ - km2.inertia_) / km1.inertia_ < 5e-3


def test_spilled_transform_streams():
    from mi355ml.preprocessing import StandardScaler
    a = _data(3000, 4, seed=3)
    S = spill_rows(torch.from_numpy(a), block_rows=700)
    sc = StandardScaler().fit(S)
    out = sc.transform(S)
    from mi355ml.core.spill import SpilledDistArray
    assert isinstance(out, SpilledDistArray)
    got = out.gather().cpu().numpy()
    ref = (a - a.mean(0)) / a.std(0)
    np.testing.assert_allclose(got, ref, rtol=1e-3, atol=1e-4)


def test_spilled_percentiles_and_robust_scaler():
    from mi355ml.ops.percentile import col_percentiles
    from mi355ml.preprocessing import RobustScaler
    a = _data(5000, 3, seed=5)
    X = from_numpy(a)
    S = spill_rows(torch.from_numpy(a), block_rows=800)
    qs = col_percentiles(S, [25.0, 50.0, 75.0]).cpu().numpy()
    ref = np.percentile(a, [25, 50, 75], axis=0)
    # per-block curves with few quantile points: coarser approximation
    # (the reference's per-chunk summaries behave the same way)
    np.testing.assert_allclose(qs, ref, atol=0.12)
    rs = RobustScaler().fit(S)
    rs2 = RobustScaler().fit(X)
    np.testing.assert_allclose(rs.center_.cpu().numpy(),
                               rs2.center_.cpu().numpy(), atol=0.12)


def _spilled_world2(rank, world, a_bytes, y_bytes, n, f):
    X = np.frombuffer(a_bytes, dtype=np.float32).reshape(n, f).copy()
    y = np.frombuffer(y_bytes, dtype=np.float32).copy()
    S = spill_rows(torch.from_numpy(X), block_rows=97)
    # streamed moments equal the resident distributed moments
    D = from_numpy(X)
    ns, ms, m2s = S.moments()
    nr, mr, m2r = D.moments()
    assert ns == nr
    np.testing.assert_allclose(ms.cpu().numpy(), mr.cpu().numpy(), rtol=1e-10)
    np.testing.assert_allclose(m2s.cpu().numpy(), m2r.cpu().numpy(),
                               rtol=1e-8)
    # spilled fit == resident fit across 2 ranks
    lr_s = LogisticRegression(max_iter=40).fit(S, from_numpy(y))
    lr_d = LogisticRegression(max_iter=40).fit(D, from_numpy(y))
    np.testing.assert_allclose(lr_s.coef_.cpu().numpy(),
                               lr_d.coef_.cpu().numpy(), rtol=1e-5,
                               atol=1e-7)


def test_spilled_world2(mp_runner):
    rng = np.random.default_rng(21)
    X = rng.standard_normal((1500, 6)).astype(np.float32)
    y = (X[:, 0] + X[:, 1] > 0).astype(np.float32)
    mp_runner(_spilled_world2, world=2,
              args=(X.tobytes(), y.tobytes(), X.shape[0], X.shape[1]))


def test_spilled_sparse_matches_resident():
    """Sparse spill tier (TODO r2): streamed-CSR dot/xt_dot/sum and a
    sparse GLM fit match the device-resident CSR results."""
    import scipy.sparse as ssp
    import torch
    from mi355ml.core.sparse import (SparseDistArray, from_scipy,
                                     spill_sparse)
    from mi355ml.linear_model import LogisticRegression
    rng = np.random.default_rng(9)
    n, dcols = 4000, 31
    dense = (rng.uniform(size=(n, dcols)) < 0.2) * \
        rng.standard_normal((n, dcols))
    S = from_scipy(ssp.csr_matrix(dense.astype(np.float32)))
    P = spill_sparse(S, block_rows=700)  # multiple partial blocks
    w = torch.from_numpy(rng.standard_normal(dcols).astype(np.float32))
    np.testing.assert_allclose(P.dot(w).gather().cpu().numpy(),
                               S.dot(w).gather().cpu().numpy(),
                               rtol=1e-5, atol=1e-5)
    r = torch.from_numpy(rng.standard_normal(n).astype(np.float32))
    np.testing.assert_allclose(P.xt_dot(r).cpu().numpy(),
                               S.xt_dot(r).cpu().numpy(), rtol=1e-10)
    np.testing.assert_allclose(P.sum().cpu().numpy(),
                               S.sum().cpu().numpy(), rtol=1e-10)
    y = (dense @ rng.standard_normal(dcols) > 0).astype(np.float32)
    m_res = LogisticRegression(solver="lbfgs", max_iter=30).fit(
        S, torch.from_numpy(y))
    m_spl = LogisticRegression(solver="lbfgs", max_iter=30).fit(
        P, torch.from_numpy(y))
    np.testing.assert_allclose(m_spl.coef_.numpy(), m_res.coef_.numpy(),
                               rtol=1e-4, atol=1e-4)


def _spilled_sparse_world2(rank, world, a_bytes, shape):
    import scipy.sparse as ssp
    import torch
    from mi355ml.core.array import from_numpy
    from mi355ml.core.sparse import from_scipy, spill_sparse
    from mi355ml.linear_model import LogisticRegression
    a = np.frombuffer(a_bytes, dtype=np.float32).reshape(shape).copy()
    S = from_scipy(ssp.csr_matrix(a))
    P = spill_sparse(S, block_rows=150)
    y = (a.sum(axis=1) > 0).astype(np.float32)
    m_spl = LogisticRegression(solver="lbfgs", max_iter=25).fit(
        P, from_numpy(y))
    m_res = LogisticRegression(solver="lbfgs", max_iter=25).fit(
        S, from_numpy(y))
    np.testing.assert_allclose(m_spl.coef_.numpy(), m_res.coef_.numpy(),
                               rtol=1e-4, atol=1e-4)


def test_spilled_sparse_world2(mp_runner):
    rng = np.random.default_rng(10)
    a = ((rng.uniform(size=(900, 13)) < 0.25) *
         rng.standard_normal((900, 13))).astype(np.float32)
    mp_runner(_spilled_sparse_world2, world=2, args=(a.tobytes(), a.shape))
