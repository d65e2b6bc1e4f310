"""Ensemble, text feature extraction, spectral clustering, utils,
diagnostics."""

import numpy as np
import scipy.sparse as sp
import torch
from sklearn.linear_model import SGDClassifier, SGDRegressor

import mi355ml
from mi355ml.core.array import from_numpy
from mi355ml.core.bag import DistList
from mi355ml.cluster import SpectralClustering
from mi355ml.ensemble import (BlockwiseVotingClassifier,
                              BlockwiseVotingRegressor)
from mi355ml.feature_extraction.text import (CountVectorizer, FeatureHasher,
                                             HashingVectorizer)


def test_blockwise_voting_classifier():
    rng = np.random.default_rng(0)
    X = rng.standard_normal((2000, 5)).astype(np.float32)
    w = rng.standard_normal(5)
    y = (X @ w > 0).astype(np.float32)
    with mi355ml.config.set({"array.block_rows": 500}):
        clf = BlockwiseVotingClassifier(
            SGDClassifier(loss="log_loss", random_state=0, max_iter=20))
        clf.fit(from_numpy(X), from_numpy(y))
        assert len(clf.estimators_) == 4
        acc = clf.score(from_numpy(X), from_numpy(y))
        assert acc > 0.8
        proba = clf.predict_proba(from_numpy(X)).gather().cpu().numpy()
        assert proba.shape == (2000, 2)


def test_blockwise_voting_regressor():
    rng = np.random.default_rng(1)
    X = rng.standard_normal((2000, 4)).astype(np.float32)
    w = rng.standard_normal(4)
    y = (X @ w).astype(np.float32)
    with mi355ml.config.set({"array.block_rows": 1000}):
        reg = BlockwiseVotingRegressor(SGDRegressor(random_state=0))
        reg.fit(from_numpy(X), from_numpy(y))
        assert reg.score(from_numpy(X), from_numpy(y)) > 0.9


DOCS = ["the quick brown fox", "jumps over the lazy dog",
        "the dog barks", "quick quick fox"]


def test_hashing_vectorizer():
    hv = HashingVectorizer(n_features=64)
    out = hv.fit_transform(DOCS)
    mats = out.gather()
    m = sp.vstack(mats)
    assert m.shape == (4, 64)


def test_count_vectorizer():
    cv = CountVectorizer()
    out = cv.fit_transform(DOCS)
    m = sp.vstack(out.gather())
    from sklearn.feature_extraction.text import CountVectorizer as SkCV
    ref = SkCV().fit(DOCS)
    assert cv.vocabulary_ == ref.vocabulary_
    np.testing.assert_array_equal(m.toarray(), ref.transform(DOCS).toarray())


def test_feature_hasher():
    items = [{"a": 1, "b": 2}, {"b": 3, "c": 1}]
    fh = FeatureHasher(n_features=32)
    m = sp.vstack(fh.transform(items).gather())
    assert m.shape == (2, 32)


def test_spectral_clustering():
    rng = np.random.default_rng(2)
    k = 3
    centers = np.array([[0, 0], [8, 8], [-8, 8]])
    lab = rng.integers(0, k, 900)
    X = (centers[lab] + rng.standard_normal((900, 2))).astype(np.float32)
    sc = SpectralClustering(n_clusters=k, n_components=60, random_state=0)
    got = sc.fit_predict(from_numpy(X)).gather().cpu().numpy()
    # clusters should align with true blobs up to permutation
    from itertools import permutations
    best = max(
        (np.mean(np.array([p[g] for g in got]) == lab)
         for p in permutations(range(k))))
    assert best > 0.9


def test_utils_svd_flip():
    from mi355ml.utils import svd_flip
    u = torch.randn(10, 3)
    v = torch.randn(3, 5)
    u2, v2 = svd_flip(u, v)
    np.testing.assert_allclose((u2 @ v2).cpu().numpy(), (u @ v).cpu().numpy(),
                               rtol=1e-5, atol=1e-5)


def test_profiler_counters():
    from mi355ml.diagnostics import Profiler, Counters, device_memory
    p = Profiler(use_events=False)
    with p.track("op1"):
        _ = sum(range(1000))
    assert "op1" in p.summary()
    c = Counters()
    c.inc("samples", 5)
    assert c.snapshot()["samples"] == 5
    device_memory()  # no GPU here: just must not raise


def test_pairwise_metrics_vs_sklearn():
    from sklearn.metrics.pairwise import (cosine_distances,
                                          manhattan_distances, rbf_kernel as
                                          sk_rbf)
    from mi355ml.metrics import (pairwise_distances, rbf_kernel,
                                 linear_kernel, polynomial_kernel)
    rng = np.random.default_rng(5)
    X = rng.standard_normal((300, 6)).astype(np.float32)
    Y = rng.standard_normal((20, 6)).astype(np.float32)
    Yt = torch.from_numpy(Y)
    got = pairwise_distances(from_numpy(X), Yt, "cosine").gather().cpu().numpy()
    np.testing.assert_allclose(got, cosine_distances(X, Y), rtol=1e-4,
                               atol=1e-5)
    got = pairwise_distances(from_numpy(X), Yt, "manhattan").gather().cpu().numpy()
    np.testing.assert_allclose(got, manhattan_distances(X, Y), rtol=1e-4,
                               atol=1e-4)
    got = rbf_kernel(from_numpy(X), Yt, gamma=0.3).gather().cpu().numpy()
    np.testing.assert_allclose(got, sk_rbf(X, Y, gamma=0.3), rtol=1e-4,
                               atol=1e-5)
    got = linear_kernel(from_numpy(X), Yt).gather().cpu().numpy()
    np.testing.assert_allclose(got, X @ Y.T, rtol=1e-4, atol=1e-4)
    got = polynomial_kernel(from_numpy(X), Yt, degree=2).gather().cpu().numpy()
    ref = (X @ Y.T / 6 + 1.0) ** 2
    np.testing.assert_allclose(got, ref, rtol=1e-3, atol=1e-4)


def test_text_pipeline_sparse_incremental():
    """HashingVectorizer -> Incremental(SGDClassifier) over sparse
    partitions — the reference's text workflow."""
    from mi355ml.wrappers import Incremental, ParallelPostFit
    docs = ["good great fine excellent", "bad awful terrible poor",
            "great good nice", "terrible bad sad", "excellent fine happy",
            "awful poor worst"] * 30
    labels = np.array([1, 0, 1, 0, 1, 0] * 30, dtype=np.int64)
    hv = HashingVectorizer(n_features=256)
    Xs = hv.transform(DistList.from_global(docs))
    ys = DistList.from_global([labels])
    inc = Incremental(SGDClassifier(loss="log_loss", random_state=0),
                      shuffle_blocks=False)
    inc.fit(Xs, ys, epochs=4)
    preds = inc.predict(Xs).gather()
    pred = np.concatenate([np.asarray(p) for p in preds])
    assert (pred == labels).mean() > 0.9


def test_xgboost_handoff_informative_error():
    import pytest as _pytest
    from mi355ml import xgboost as mx
    from mi355ml.core.array import from_numpy
    X = from_numpy(np.zeros((10, 2), dtype=np.float32))
    y = from_numpy(np.zeros(10, dtype=np.float32))
    with _pytest.raises(ImportError, match="xgboost"):
        mx.train({}, X, y)


def test_metrics_match_sklearn():
    import sklearn.metrics as skm
    from mi355ml.metrics import (accuracy_score, log_loss,
                                 mean_absolute_error, mean_squared_error,
                                 r2_score)
    rng = np.random.default_rng(29)
    yt = rng.integers(0, 2, 2000).astype(np.float32)
    yp = rng.integers(0, 2, 2000).astype(np.float32)
    proba = rng.uniform(0.01, 0.99, 2000).astype(np.float32)
    p2 = np.stack([1 - proba, proba], axis=1)
    from mi355ml.core.array import from_numpy
    assert abs(float(accuracy_score(from_numpy(yt), from_numpy(yp)))
               - skm.accuracy_score(yt, yp)) < 1e-6
    assert abs(float(log_loss(from_numpy(yt), from_numpy(p2)))
               - skm.log_loss(yt, p2)) < 1e-5
    yr = rng.standard_normal(2000).astype(np.float32)
    pr = (yr + 0.3 * rng.standard_normal(2000)).astype(np.float32)
    for ours, ref in ((mean_squared_error, skm.mean_squared_error),
                      (mean_absolute_error, skm.mean_absolute_error),
                      (r2_score, skm.r2_score)):
        assert abs(float(ours(from_numpy(yr), from_numpy(pr)))
                   - ref(yr, pr)) < 1e-4


def test_r2_constant_y_true_edge_cases():
    # sklearn: perfect prediction of a constant y -> 1.0, otherwise 0.0
    from mi355ml.metrics import r2_score
    y = np.full(64, 3.0, dtype=np.float32)
    assert r2_score(from_numpy(y), from_numpy(y.copy())) == 1.0
    assert r2_score(from_numpy(y), from_numpy(y + 1.0)) == 0.0


def test_utils_functions():
    from mi355ml.utils import (check_chunks, draw_seed,
                               handle_zeros_in_scale, svd_flip)
    import torch
    # svd_flip: deterministic signs, consistent u/v flip
    u = torch.tensor([[1.0, -2.0], [3.0, 4.0]])
    v = torch.tensor([[-1.0, 2.0], [3.0, -4.0]])
    uf, vf = svd_flip(u.clone(), v.clone())
    uf2, vf2 = svd_flip(u.clone(), v.clone())
    assert torch.equal(uf, uf2) and torch.equal(vf, vf2)
    torch.testing.assert_close(uf @ vf, u @ v)  # product preserved
    # zeros in scale replaced by 1 (sklearn semantics)
    s = handle_zeros_in_scale(torch.tensor([0.0, 2.0, 0.0]))
    assert s.tolist() == [1.0, 2.0, 1.0]
    # chunks tuple sane
    r, c = check_chunks(100, 10)
    assert r >= 1 and c >= 1
    # draw_seed in range + deterministic per rng state
    rng = np.random.default_rng(0)
    a = draw_seed(np.random.default_rng(0))
    b = draw_seed(np.random.default_rng(0))
    assert a == b and 0 <= a < 2 ** 31


def test_diagnostics_profiler_counters():
    from mi355ml.diagnostics import Counters, Profiler, device_memory
    prof = Profiler(use_events=False)
    with prof.track("opA", rows=10):
        pass
    with prof.track("opA"):
        pass
    with prof.track("opB"):
        pass
    s = prof.summary()
    assert s["opA"]["n"] == 2 and s["opB"]["n"] == 1
    assert "opA" in prof.report()
    c = Counters()
    c.inc("x.calls")
    c.inc("x.bytes", 128.0)
    c.inc("x.calls")
    snap = c.snapshot()
    assert snap["x.calls"] == 2 and snap["x.bytes"] == 128.0
    dm = device_memory()
    assert isinstance(dm, dict)
