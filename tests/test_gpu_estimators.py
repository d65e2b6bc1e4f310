"""GPU end-to-end tests for the wider estimator surface (all marked gpu).
These exercise the HIP ops + rocSOLVER/rocBLAS library paths on device."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


def test_pca_gpu():
    from mi355ml import datasets
    from mi355ml.decomposition import PCA
    X, _ = datasets.make_regression(n_samples=200_000, n_features=50,
                                    random_state=0)
    assert X.device.type == "cuda"
    pca = PCA(n_components=10, svd_solver="randomized").fit(X)
    assert pca.components_.shape == (10, 50)
    Z = pca.transform(X)
    assert Z.shape == (200_000, 10)
    r = pca.explained_variance_ratio_.sum()
    assert 0.2 < float(r) <= 1.001


def test_tsvd_gpu_tsqr():
    from mi355ml import datasets
    from mi355ml.decomposition import TruncatedSVD
    X, _ = datasets.make_regression(n_samples=100_000, n_features=200,
                                    random_state=1)
    t = TruncatedSVD(n_components=20, algorithm="tsqr").fit(X)
    assert t.singular_values_.shape == (20,)
    assert bool((t.singular_values_[:-1] >= t.singular_values_[1:]).all())


def test_robust_scaler_gpu():
    from mi355ml import datasets
    from mi355ml.preprocessing import RobustScaler
    X, _ = datasets.make_regression(n_samples=300_000, n_features=20,
                                    random_state=2)
    rs = RobustScaler().fit(X)
    out = rs.transform(X)
    med = out.gather().median(dim=0).values
    assert float(med.abs().max()) < 0.05


def test_hyperband_gpu_data():
    from sklearn.linear_model import SGDClassifier
    from mi355ml import datasets
    from mi355ml.model_selection import HyperbandSearchCV
    X, y = datasets.make_classification(n_samples=50_000, n_features=20,
                                        n_informative=5, random_state=3)
    hb = HyperbandSearchCV(
        SGDClassifier(loss="log_loss", random_state=0),
        {"alpha": [1e-5, 1e-4, 1e-3]}, max_iter=4, random_state=0)
    hb.fit(X, y)
    assert hb.best_score_ > 0.6


def test_glm_families_gpu_e2e():
    from mi355ml import datasets
    from mi355ml.linear_model import LinearRegression, PoissonRegression
    X, y = datasets.make_regression(n_samples=500_000, n_features=50,
                                    noise=0.1, random_state=4)
    lin = LinearRegression(C=1e8, max_iter=200).fit(X, y)
    assert lin.score(X, y) > 0.99
    Xp, yp = datasets.make_counts(n_samples=200_000, n_features=20,
                                  random_state=5)
    p = PoissonRegression(C=1e6, max_iter=100).fit(Xp, yp)
    assert p.n_iter_ >= 1


def test_kmeans_gpu_matches_cpu_one_iter():
    """Bitwise-comparable single Lloyd iteration: GPU fused kernel partials
    vs CPU torch oracle on the same data."""
    from mi355ml.ops import kmeans_assign
    g = torch.Generator().manual_seed(7)
    x = torch.randn(200_000, 50, generator=g)
    c = torch.randn(64, 50, generator=g)
    lab_c, sums_c, cnt_c, in_c = kmeans_assign(x, c)
    lab_g, sums_g, cnt_g, in_g = kmeans_assign(x.cuda(), c.cuda())
    new_c = sums_c / cnt_c.clamp(min=1)[:, None]
    new_g = sums_g.cpu() / cnt_g.cpu().clamp(min=1)[:, None]
    torch.testing.assert_close(new_g, new_c, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(in_g.cpu(), in_c, rtol=1e-5, atol=1e-2)


def test_spill_gpu_streams_blocks():
    import torch
    from mi355ml.core.spill import spill_rows
    from mi355ml.core.array import shard_rows
    from mi355ml.linear_model import LogisticRegression
    g = torch.Generator().manual_seed(11)
    a = torch.randn(2_000_000, 50, generator=g)
    w = torch.randn(50, generator=g)
    y = (torch.sigmoid(a @ w) > torch.rand(2_000_000, generator=g)).float()
    S = spill_rows(a, block_rows=500_000)   # host-pinned shard
    assert S.local.device.type == "cpu" and S.device.type == "cuda"
    lr = LogisticRegression(C=1.0, max_iter=50).fit(S, shard_rows(y))
    lr2 = LogisticRegression(C=1.0, max_iter=50).fit(shard_rows(a),
                                                     shard_rows(y))
    torch.testing.assert_close(lr.coef_.cpu(), lr2.coef_.cpu(), rtol=1e-3,
                               atol=1e-4)


def test_kmeans_gpu_bitwise_deterministic():
    """Run-to-run determinism: no atomics in the fused path."""
    from mi355ml.ops import kmeans_assign
    import torch
    g = torch.Generator().manual_seed(13)
    x = torch.randn(1_000_000, 50, generator=g).cuda()
    c = torch.randn(100, 50, generator=g).cuda()
    l1, s1, c1, i1 = kmeans_assign(x, c)
    l2, s2, c2, i2 = kmeans_assign(x, c)
    assert torch.equal(l1, l2)
    assert torch.equal(s1, s2)
    assert torch.equal(c1, c2)
    assert torch.equal(i1, i2)


def test_tdigest_gpu():
    from mi355ml.core.array import from_numpy
    from mi355ml.ops.percentile import col_percentiles
    rng = np.random.default_rng(11)
    a = rng.standard_normal((500_000, 8)).astype(np.float32)
    X = from_numpy(a)
    assert X.device.type == "cuda"
    qs = [1.0, 25.0, 50.0, 75.0, 99.0]
    td = col_percentiles(X, qs, method="tdigest").cpu().numpy()
    ex = np.percentile(a, qs, axis=0)
    assert np.abs(td - ex).max() < 0.01


def test_sparse_onehot_logreg_gpu():
    import scipy.sparse as sp
    from mi355ml.core.array import from_numpy
    from mi355ml.core.sparse import SparseDistArray
    from mi355ml.linear_model import LogisticRegression
    from mi355ml.preprocessing import OneHotEncoder
    rng = np.random.default_rng(12)
    a = rng.integers(0, 8, (200_000, 5)).astype(np.float32)
    y = (a[:, 0] + a[:, 1] > 7).astype(np.float32)
    enc = OneHotEncoder(sparse_output=True).fit(from_numpy(a))
    S = enc.transform(from_numpy(a))
    assert isinstance(S, SparseDistArray) and S.device.type == "cuda"
    lr = LogisticRegression(C=1.0, max_iter=60).fit(S, from_numpy(y))
    acc = float((lr.predict(S).gather().cpu().numpy() == y).mean())
    assert acc > 0.9
    # sparse fit == dense fit on device
    lrd = LogisticRegression(C=1.0, max_iter=60).fit(S.todense(),
                                                     from_numpy(y))
    np.testing.assert_allclose(lr.coef_.cpu().numpy(),
                               lrd.coef_.cpu().numpy(), rtol=1e-3, atol=1e-3)


def test_sparse_xtdot_bitwise_deterministic_gpu():
    import scipy.sparse as sp
    from mi355ml.core.sparse import from_scipy
    rng = np.random.default_rng(13)
    M = sp.random(2_000_000, 300, density=0.02, random_state=1,
                  format="csr", dtype=np.float32)
    S = from_scipy(M)
    r = torch.randn(2_000_000, device="cuda", generator=torch.Generator(
        device="cuda").manual_seed(0))
    g1 = S.xt_dot(r).cpu().numpy()
    g2 = S.xt_dot(r).cpu().numpy()
    assert (g1 == g2).all()  # ordered segment sums, no atomics
    z1 = S.dot(torch.ones(300, device="cuda")).gather().cpu().numpy()
    z2 = S.dot(torch.ones(300, device="cuda")).gather().cpu().numpy()
    assert (z1 == z2).all()


def test_kmeans_large_k_fallback_deterministic_gpu():
    from mi355ml.ops.kmeans import kmeans_assign
    g = torch.Generator(device="cuda").manual_seed(7)
    x = torch.randn(3_000_000, 32, generator=g, device="cuda")
    c = torch.randn(400, 32, generator=g, device="cuda")  # k>256: composed
    l1, s1, n1, i1 = kmeans_assign(x, c)
    l2, s2, n2, i2 = kmeans_assign(x, c)
    assert (l1 == l2).all()
    assert (s1 == s2).all() and (n1 == n2).all()
    assert float(i1) == float(i2)


def test_kmeans_fit_uses_split_kernel():
    """The production Lloyd path must be the wave-specialized split
    kernel, not a fallback (native-loaded evidence for the driver)."""
    import torch
    from mi355ml.cluster import KMeans
    from mi355ml.core.array import from_numpy
    from mi355ml.ops import kmeans as km_mod
    import numpy as np
    rng = np.random.default_rng(0)
    # the Lloyd op itself must take the wave-specialized split kernel
    for d in (100, 50):  # 64-row and 128-row variants
        x = torch.from_numpy(
            rng.standard_normal((200_000, d)).astype(np.float32)).cuda()
        c = torch.from_numpy(
            rng.standard_normal((100, d)).astype(np.float32)).cuda()
        km_mod.kmeans_assign(x, c)
        assert km_mod.last_path == "fused-split", (d, km_mod.last_path)
    # end-to-end fit stays on a HIP path throughout (the final labels
    # pass is closest_center -> "fused")
    X = rng.standard_normal((200_000, 100)).astype(np.float32)
    KMeans(n_clusters=100, max_iter=3, random_state=0).fit(from_numpy(X))
    assert km_mod.last_path in ("fused-split", "fused"), km_mod.last_path


def test_spilled_sparse_gpu():
    """Sparse spill tier streams CSR blocks through the copy stream on
    device; fit matches the resident-sparse fit."""
    import scipy.sparse as ssp
    import torch
    from mi355ml.core.array import from_numpy
    from mi355ml.core.sparse import from_scipy, spill_sparse
    from mi355ml.linear_model import LogisticRegression
    rng = np.random.default_rng(12)
    a = ((rng.uniform(size=(200_000, 64)) < 0.1) *
         rng.standard_normal((200_000, 64))).astype(np.float32)
    S = from_scipy(ssp.csr_matrix(a))
    P = spill_sparse(S, block_rows=30_000)
    y = (a.sum(axis=1) > 0).astype(np.float32)
    m_spl = LogisticRegression(solver="lbfgs", max_iter=20).fit(
        P, from_numpy(y))
    m_res = LogisticRegression(solver="lbfgs", max_iter=20).fit(
        S, from_numpy(y))
    torch.testing.assert_close(m_spl.coef_, m_res.coef_, rtol=1e-4,
                               atol=1e-4)
