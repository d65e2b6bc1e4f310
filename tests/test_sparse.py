"""SparseDistArray (CSR shards): encoder sparse output + sparse GLM path
(dask-ml sparse_output=True parity, scipy/sklearn oracles)."""

import numpy as np
import scipy.sparse as sp
import pytest
import torch
from sklearn.preprocessing import OneHotEncoder as SkOneHot

from mi355ml.core.array import from_numpy
from mi355ml.core.sparse import SparseDistArray, from_scipy
from mi355ml.preprocessing import OneHotEncoder


def _cat_data(n=4000, f=3, k=7, seed=0):
    rng = np.random.default_rng(seed)
    return rng.integers(0, k, (n, f)).astype(np.float32)


def test_onehot_sparse_matches_dense_and_sklearn():
    a = _cat_data()
    enc = OneHotEncoder(sparse_output=True).fit(from_numpy(a))
    S = enc.transform(from_numpy(a))
    assert isinstance(S, SparseDistArray)
    dense = enc.transform(from_numpy(a)).todense().gather().cpu().numpy()
    enc_d = OneHotEncoder(sparse_output=False).fit(from_numpy(a))
    ref_dense = enc_d.transform(from_numpy(a)).gather().cpu().numpy()
    np.testing.assert_array_equal(dense, ref_dense)
    sk = SkOneHot(sparse_output=True).fit(a)
    got = S.gather()
    assert sp.issparse(got)
    np.testing.assert_array_equal(got.toarray(),
                                  sk.transform(a).toarray())
    # nnz = one per (row, feature)
    assert S.nnz == a.shape[0] * a.shape[1]


def test_sparse_dot_xtdot_sum():
    rng = np.random.default_rng(1)
    M = sp.random(500, 40, density=0.1, random_state=2, format="csr",
                  dtype=np.float64)
    S = from_scipy(M)
    w = rng.standard_normal(40)
    z = S.dot(torch.from_numpy(w)).gather().cpu().numpy()
    np.testing.assert_allclose(z, M @ w, rtol=1e-12)
    r = rng.standard_normal(500)
    g = S.xt_dot(torch.from_numpy(r))
    np.testing.assert_allclose(g.cpu().numpy(), M.T @ r, rtol=1e-10)
    np.testing.assert_allclose(S.sum().cpu().numpy(),
                               np.asarray(M.sum(axis=0)).ravel(), rtol=1e-12)
    # roundtrip
    np.testing.assert_array_equal(S.gather().toarray(), M.toarray())


def test_sparse_logreg_matches_dense():
    rng = np.random.default_rng(3)
    a = _cat_data(3000, 4, 5, seed=3)
    y = (a[:, 0] + a[:, 1] > 4).astype(np.float32)
    enc = OneHotEncoder(sparse_output=True).fit(from_numpy(a))
    S = enc.transform(from_numpy(a))
    D = S.todense()
    from mi355ml.linear_model import LogisticRegression
    lrs = LogisticRegression(C=1.0, max_iter=100).fit(S, from_numpy(y))
    lrd = LogisticRegression(C=1.0, max_iter=100).fit(D, from_numpy(y))
    np.testing.assert_allclose(lrs.coef_.cpu().numpy(),
                               lrd.coef_.cpu().numpy(), rtol=1e-3, atol=1e-3)
    # sparse predict path
    ps = lrs.predict(S).gather().cpu().numpy()
    pd_ = lrd.predict(D).gather().cpu().numpy()
    assert (ps == pd_).mean() > 0.99


def _sparse_world2(rank, world, a_bytes, shape):
    a = np.frombuffer(a_bytes, dtype=np.float32).reshape(shape).copy()
    enc = OneHotEncoder(sparse_output=True).fit(from_numpy(a))
    S = enc.transform(from_numpy(a))
    sk = SkOneHot(sparse_output=True).fit(a)
    np.testing.assert_array_equal(S.gather().toarray(),
                                  sk.transform(a).toarray())
    g = S.sum().cpu().numpy()
    np.testing.assert_allclose(
        g, np.asarray(sk.transform(a).sum(axis=0)).ravel(), rtol=1e-10)


def test_sparse_world2(mp_runner):
    a = _cat_data(800, 2, 4, seed=4)
    mp_runner(_sparse_world2, world=2, args=(a.tobytes(), a.shape))


def test_sparse_multivector_dot():
    rng = np.random.default_rng(5)
    M = sp.random(300, 25, density=0.15, random_state=6, format="csr",
                  dtype=np.float64)
    S = from_scipy(M)
    W = rng.standard_normal((25, 4))
    Z = S.dot(torch.from_numpy(W)).gather().cpu().numpy()
    np.testing.assert_allclose(Z, M @ W, rtol=1e-12)
    with pytest.raises(NotImplementedError):
        S.sum(axis=1)


def _sparse_rebalance_world2(rank, world, a_bytes, shape):
    import scipy.sparse as ssp
    import torch
    from mi355ml.core.sparse import SparseDistArray
    from mi355ml.parallel import comm as comm_mod
    a = np.frombuffer(a_bytes, dtype=np.float32).reshape(shape).copy()
    comm = comm_mod.default()
    # deliberately UNEVEN shards: rank 0 gets 10 rows, rank 1 the rest
    split = 10
    lo, hi = (0, split) if rank == 0 else (split, shape[0])
    local = ssp.csr_matrix(a[lo:hi])
    S = SparseDistArray(torch.from_numpy(local.indptr.astype(np.int64)),
                        torch.from_numpy(local.indices.astype(np.int64)),
                        torch.from_numpy(local.data), shape[1],
                        shape[0], lo, comm)
    B = S.rebalance()
    # balanced: row counts differ by at most 1
    nl = comm.all_gather_object(B.local_rows)
    assert max(nl) - min(nl) <= 1
    assert sum(nl) == shape[0]
    # content identical to the original global matrix
    np.testing.assert_allclose(B.gather().toarray(), a, rtol=0, atol=0)
    # offsets are the canonical contiguous layout
    offs = comm.all_gather_object(B.row_offset)
    assert offs == sorted(offs)
    # sparse ops still work post-rebalance
    w = np.random.default_rng(0).standard_normal(shape[1]).astype(np.float32)
    z = B.dot(torch.from_numpy(w)).gather().cpu().numpy()
    np.testing.assert_allclose(z, a @ w, rtol=1e-4, atol=1e-4)


def test_sparse_rebalance_world2(mp_runner):
    rng = np.random.default_rng(11)
    a = (rng.uniform(size=(101, 7)) < 0.3).astype(np.float32) * \
        rng.standard_normal((101, 7)).astype(np.float32)
    mp_runner(_sparse_rebalance_world2, world=2, args=(a.tobytes(), a.shape))
