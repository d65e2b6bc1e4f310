"""PCA / TruncatedSVD / IncrementalPCA vs sklearn oracle."""

import numpy as np
import pytest
from sklearn.decomposition import PCA as SkPCA
from sklearn.decomposition import TruncatedSVD as SkTSVD

from mi355ml.core.array import from_numpy
from mi355ml.decomposition import PCA, TruncatedSVD, IncrementalPCA


def _lowrank(n=2000, d=20, r=5, noise=0.01, seed=0):
    rng = np.random.default_rng(seed)
    A = rng.standard_normal((n, r))
    B = rng.standard_normal((r, d)) * np.linspace(5, 1, r)[:, None]
    X = A @ B + rng.standard_normal((n, d)) * noise + rng.uniform(-1, 1, d)
    return X.astype(np.float32)


def _align_signs(got, ref):
    # components are sign-ambiguous; align each row before comparing
    signs = np.sign((got * ref).sum(axis=1))
    signs[signs == 0] = 1
    return got * signs[:, None]


@pytest.mark.parametrize("solver", ["full", "randomized"])
def test_pca_matches_sklearn(solver):
    X = _lowrank()
    ours = PCA(n_components=5, svd_solver=solver).fit(from_numpy(X))
    ref = SkPCA(n_components=5, svd_solver="full").fit(X)
    np.testing.assert_allclose(ours.mean_.cpu().numpy(), ref.mean_, rtol=1e-4,
                               atol=1e-4)
    np.testing.assert_allclose(ours.singular_values_.cpu().numpy(),
                               ref.singular_values_, rtol=1e-3)
    got = _align_signs(ours.components_.cpu().numpy(), ref.components_)
    np.testing.assert_allclose(got, ref.components_, rtol=5e-2, atol=5e-3)
    np.testing.assert_allclose(ours.explained_variance_ratio_.cpu().numpy(),
                               ref.explained_variance_ratio_, rtol=1e-2,
                               atol=1e-4)


def test_pca_transform_roundtrip():
    X = _lowrank(seed=1)
    pca = PCA(n_components=5).fit(from_numpy(X))
    Z = pca.transform(from_numpy(X))
    assert Z.shape == (X.shape[0], 5)
    back = pca.inverse_transform(Z).gather().cpu().numpy()
    # low-rank + small noise: reconstruction error should be tiny
    err = np.abs(back - X).mean() / np.abs(X).mean()
    assert err < 0.02
    ft = pca.fit_transform(from_numpy(X)).gather().cpu().numpy()
    np.testing.assert_allclose(np.abs(ft), np.abs(Z.gather().cpu().numpy()),
                               rtol=1e-2, atol=1e-2)


def test_tsvd_matches_sklearn():
    X = _lowrank(seed=2)
    ours = TruncatedSVD(n_components=4, algorithm="tsqr").fit(from_numpy(X))
    ref = SkTSVD(n_components=4, algorithm="arpack").fit(X.astype(np.float64))
    np.testing.assert_allclose(ours.singular_values_.cpu().numpy(),
                               ref.singular_values_, rtol=1e-3)
    got = _align_signs(ours.components_.cpu().numpy(), ref.components_)
    np.testing.assert_allclose(got, ref.components_, rtol=5e-2, atol=5e-3)


def test_tsvd_randomized_reconstruction():
    # the column offset adds a rank-1 term: the matrix is effectively rank 6
    X = _lowrank(seed=3, noise=0.001)
    t = TruncatedSVD(n_components=6, algorithm="randomized").fit(
        from_numpy(X))
    Z = t.transform(from_numpy(X))
    back = t.inverse_transform(Z).gather().cpu().numpy()
    err = np.linalg.norm(back - X) / np.linalg.norm(X)
    assert err < 0.02


def test_incremental_pca():
    X = _lowrank(seed=4)
    ours = IncrementalPCA(n_components=5, batch_size=300).fit(from_numpy(X))
    ref = SkPCA(n_components=5).fit(X)
    np.testing.assert_allclose(ours.singular_values_.cpu().numpy(),
                               ref.singular_values_, rtol=5e-2)
    got = _align_signs(ours.components_.cpu().numpy(), ref.components_)
    np.testing.assert_allclose(np.abs(got), np.abs(ref.components_),
                               rtol=0.2, atol=0.05)
    Z = ours.transform(from_numpy(X)).gather().cpu().numpy()
    assert Z.shape == (X.shape[0], 5)


def _pca_world2(rank, world, X_bytes, n, d, sv_ref):
    X = np.frombuffer(X_bytes, dtype=np.float32).reshape(n, d).copy()
    ours = PCA(n_components=5, svd_solver="full").fit(from_numpy(X))
    np.testing.assert_allclose(ours.singular_values_.cpu().numpy(), sv_ref,
                               rtol=1e-3)


def test_pca_world2(mp_runner):
    X = _lowrank(n=1501, seed=5)
    ref = SkPCA(n_components=5, svd_solver="full").fit(X)
    mp_runner(_pca_world2, world=2,
              args=(X.tobytes(), X.shape[0], X.shape[1],
                    ref.singular_values_))


def test_pca_whiten_matches_sklearn():
    from sklearn.decomposition import PCA as SkPCA
    rng = np.random.default_rng(17)
    B = rng.standard_normal((5, 12))
    X = (rng.standard_normal((4000, 5)) @ B
         + 0.05 * rng.standard_normal((4000, 12))).astype(np.float32)
    ours = PCA(n_components=4, whiten=True, svd_solver="full").fit(
        from_numpy(X))
    ref = SkPCA(n_components=4, whiten=True).fit(X)
    Z = ours.transform(from_numpy(X)).gather().cpu().numpy()
    Zr = ref.transform(X)
    # whitened components have unit variance; sign per component is free
    np.testing.assert_allclose(Z.std(axis=0), Zr.std(axis=0), rtol=1e-2)
    cos = np.abs((Z * Zr).mean(axis=0) / (Z.std(axis=0) * Zr.std(axis=0)))
    assert (cos > 0.99).all()
    # inverse_transform roundtrip
    back = ours.inverse_transform(from_numpy(Z)).gather().cpu().numpy()
    np.testing.assert_allclose(back, ref.inverse_transform(Zr), rtol=1e-2,
                               atol=1e-2)
