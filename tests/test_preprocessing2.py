"""RobustScaler / QuantileTransformer / encoders / imputer / GaussianNB
vs sklearn oracles."""

import numpy as np
import pytest
import torch
from sklearn.impute import SimpleImputer as SkImputer
from sklearn.naive_bayes import GaussianNB as SkGNB
from sklearn.preprocessing import OneHotEncoder as SkOneHot
from sklearn.preprocessing import OrdinalEncoder as SkOrdinal
from sklearn.preprocessing import PolynomialFeatures as SkPoly
from sklearn.preprocessing import RobustScaler as SkRobust

from mi355ml.core.array import from_numpy
from mi355ml.impute import SimpleImputer
from mi355ml.naive_bayes import GaussianNB
from mi355ml.preprocessing import (BlockTransformer, LabelEncoder,
                                   Normalizer, OneHotEncoder, OrdinalEncoder,
                                   PolynomialFeatures, QuantileTransformer,
                                   RobustScaler)


def _data(n=5000, f=5, seed=0):
    rng = np.random.default_rng(seed)
    return (rng.standard_normal((n, f)) * rng.uniform(1, 4, f)
            + rng.uniform(-2, 2, f)).astype(np.float32)


def test_robust_scaler():
    a = _data()
    ours = RobustScaler().fit(from_numpy(a))
    ref = SkRobust().fit(a)
    np.testing.assert_allclose(ours.center_.cpu().numpy(), ref.center_, rtol=0.05,
                               atol=0.05)
    np.testing.assert_allclose(ours.scale_.cpu().numpy(), ref.scale_, rtol=0.05)
    got = ours.transform(from_numpy(a)).gather().cpu().numpy()
    np.testing.assert_allclose(got, ref.transform(a), rtol=0.1, atol=0.1)
    back = ours.inverse_transform(from_numpy(got)).gather().cpu().numpy()
    np.testing.assert_allclose(back, a, rtol=1e-3, atol=1e-3)


def test_quantile_transformer_uniform():
    a = _data(seed=1)
    qt = QuantileTransformer(n_quantiles=200).fit(from_numpy(a))
    got = qt.transform(from_numpy(a)).gather().cpu().numpy()
    assert got.min() >= 0 and got.max() <= 1
    # transformed columns should be ~uniform: mean ~0.5, var ~1/12
    assert np.allclose(got.mean(axis=0), 0.5, atol=0.02)
    assert np.allclose(got.var(axis=0), 1 / 12, atol=0.01)
    back = qt.inverse_transform(from_numpy(got)).gather().cpu().numpy()
    np.testing.assert_allclose(back, a, rtol=0.05, atol=0.05)


def test_quantile_transformer_normal():
    a = _data(seed=2)
    qt = QuantileTransformer(n_quantiles=200,
                             output_distribution="normal").fit(from_numpy(a))
    got = qt.transform(from_numpy(a)).gather().cpu().numpy()
    assert np.allclose(got.mean(axis=0), 0.0, atol=0.05)
    assert np.allclose(got.std(axis=0), 1.0, atol=0.1)


def test_label_encoder():
    rng = np.random.default_rng(3)
    y = rng.choice([3.0, 7.0, 11.0], size=1000).astype(np.float32)
    le = LabelEncoder().fit(from_numpy(y))
    np.testing.assert_array_equal(le.classes_.cpu().numpy(), [3.0, 7.0, 11.0])
    t = le.transform(from_numpy(y)).gather().cpu().numpy()
    assert set(np.unique(t)) == {0, 1, 2}
    back = le.inverse_transform(from_numpy(t.astype(np.int64))).gather().cpu().numpy()
    np.testing.assert_array_equal(back, y)


def test_ordinal_and_onehot():
    rng = np.random.default_rng(4)
    X = rng.choice([0.0, 2.0, 5.0], size=(500, 3)).astype(np.float32)
    oe = OrdinalEncoder().fit(from_numpy(X))
    ref = SkOrdinal().fit(X)
    got = oe.transform(from_numpy(X)).gather().cpu().numpy()
    np.testing.assert_array_equal(got, ref.transform(X))
    oh = OneHotEncoder().fit(from_numpy(X))
    refh = SkOneHot(sparse_output=False).fit(X)
    goth = oh.transform(from_numpy(X)).gather().cpu().numpy()
    np.testing.assert_array_equal(goth, refh.transform(X))


def test_polynomial_features():
    a = _data(n=200, f=3, seed=5)
    pf = PolynomialFeatures(degree=2).fit(from_numpy(a))
    ref = SkPoly(degree=2).fit(a)
    got = pf.transform(from_numpy(a)).gather().cpu().numpy()
    np.testing.assert_allclose(got, ref.transform(a), rtol=1e-4, atol=1e-4)
    assert list(pf.get_feature_names_out()) == list(
        ref.get_feature_names_out(["x0", "x1", "x2"]))


def test_block_transformer():
    a = _data(n=100, f=2, seed=6)
    bt = BlockTransformer(lambda t: t * 2.0)
    got = bt.fit_transform(from_numpy(a)).gather().cpu().numpy()
    np.testing.assert_allclose(got, a * 2.0)


def test_normalizer():
    a = _data(n=100, f=4, seed=7)
    nm = Normalizer().fit_transform(from_numpy(a)).gather().cpu().numpy()
    np.testing.assert_allclose(np.linalg.norm(nm, axis=1), 1.0, rtol=1e-5)


@pytest.mark.parametrize("strategy", ["mean", "median", "most_frequent",
                                      "constant"])
def test_simple_imputer(strategy):
    rng = np.random.default_rng(8)
    a = rng.choice([1.0, 2.0, 2.0, 5.0, np.nan], size=(800, 3)).astype(
        np.float32)
    kw = {"fill_value": -1.0} if strategy == "constant" else {}
    ours = SimpleImputer(strategy=strategy, **kw).fit(from_numpy(a))
    ref = SkImputer(strategy=strategy, **kw).fit(a)
    np.testing.assert_allclose(ours.statistics_.cpu().numpy(), ref.statistics_,
                               rtol=0.06, atol=0.06)
    got = ours.transform(from_numpy(a)).gather().cpu().numpy()
    assert not np.isnan(got).any()


def test_gaussian_nb():
    rng = np.random.default_rng(9)
    n = 3000
    y = rng.integers(0, 3, n)
    means = np.array([[0, 0], [3, 3], [-3, 3]])
    X = (means[y] + rng.standard_normal((n, 2))).astype(np.float32)
    ours = GaussianNB().fit(from_numpy(X), from_numpy(y.astype(np.float32)))
    ref = SkGNB().fit(X, y)
    np.testing.assert_allclose(ours.theta_.cpu().numpy(), ref.theta_, rtol=1e-3,
                               atol=1e-3)
    np.testing.assert_allclose(ours.var_.cpu().numpy(), ref.var_, rtol=1e-2)
    acc = ours.score(from_numpy(X), from_numpy(y.astype(np.float32)))
    assert abs(acc - ref.score(X, y)) < 0.01
    proba = ours.predict_proba(from_numpy(X)).gather().cpu().numpy()
    np.testing.assert_allclose(proba.sum(axis=1), 1.0, rtol=1e-5)


# -- t-digest percentile sketch (method='tdigest' parity) -------------------

def test_tdigest_percentiles_accuracy():
    rng = np.random.default_rng(3)
    a = np.stack([rng.standard_normal(100_000),
                  rng.exponential(3.0, 100_000),
                  rng.uniform(-5, 5, 100_000)], axis=1).astype(np.float32)
    from mi355ml.ops.percentile import col_percentiles
    qs = [0.1, 1, 25, 50, 75, 99, 99.9]
    td = col_percentiles(from_numpy(a), qs, method="tdigest").cpu().numpy()
    # rank accuracy: each estimate must land within 0.1 percentile points
    for c in range(a.shape[1]):
        col = np.sort(a[:, c])
        ranks = np.searchsorted(col, td[:, c]) / len(col) * 100
        assert np.abs(ranks - np.asarray(qs)).max() < 0.1
    # monotone in q, exact at the endpoints
    assert (np.diff(td, axis=0) >= 0).all()
    ends = col_percentiles(from_numpy(a), [0.0, 100.0],
                           method="tdigest").cpu().numpy()
    np.testing.assert_allclose(ends[0], a.min(axis=0), rtol=1e-6)
    np.testing.assert_allclose(ends[1], a.max(axis=0), rtol=1e-6)


def test_tdigest_nan_and_scalers():
    rng = np.random.default_rng(4)
    a = _data(20_000, 3, seed=4)
    a[rng.integers(0, len(a), 300), 1] = np.nan
    from mi355ml.ops.percentile import col_percentiles
    td = col_percentiles(from_numpy(a), [50.0], method="tdigest")
    med = np.nanmedian(a, axis=0)
    np.testing.assert_allclose(td[0].cpu().numpy(), med, atol=0.05)
    # RobustScaler(method='tdigest') agrees with sklearn
    b = _data(seed=6)
    ours = RobustScaler(method="tdigest").fit(from_numpy(b))
    ref = SkRobust().fit(b)
    np.testing.assert_allclose(ours.center_.cpu().numpy(), ref.center_,
                               rtol=0.05, atol=0.05)
    np.testing.assert_allclose(ours.scale_.cpu().numpy(), ref.scale_,
                               rtol=0.05)
    qt = QuantileTransformer(n_quantiles=200,
                             method="tdigest").fit(from_numpy(b))
    got = qt.transform(from_numpy(b)).gather().cpu().numpy()
    assert np.allclose(got.mean(axis=0), 0.5, atol=0.02)


def _tdigest_world2(rank, world, a_bytes, shape):
    from mi355ml.ops.percentile import col_percentiles
    a = np.frombuffer(a_bytes, dtype=np.float32).reshape(shape).copy()
    qs = [5.0, 50.0, 95.0]
    td = col_percentiles(from_numpy(a), qs, method="tdigest").cpu().numpy()
    ex = np.percentile(a, qs, axis=0)
    spread = a.max(axis=0) - a.min(axis=0)
    assert (np.abs(td - ex) / spread).max() < 0.01


def test_tdigest_world2(mp_runner):
    a = _data(30_000, 4, seed=7)
    mp_runner(_tdigest_world2, world=2, args=(a.tobytes(), a.shape))


def test_onehot_drop_first():
    rng = np.random.default_rng(8)
    a = rng.integers(0, 5, (2000, 3)).astype(np.float32)
    ours = OneHotEncoder(drop="first").fit(from_numpy(a))
    ref = SkOneHot(drop="first", sparse_output=False).fit(a)
    got = ours.transform(from_numpy(a)).gather().cpu().numpy()
    np.testing.assert_array_equal(got, ref.transform(a))
    assert len(ours.get_feature_names_out()) == got.shape[1]
    # sparse drop path agrees with dense
    sp_got = OneHotEncoder(drop="first", sparse_output=True).fit(
        from_numpy(a)).transform(from_numpy(a)).gather().toarray()
    np.testing.assert_array_equal(sp_got, got)


def test_split_no_shuffle_and_kfold_shuffle():
    from mi355ml.model_selection import train_test_split, KFold
    a = np.arange(100, dtype=np.float32)[:, None]
    Xtr, Xte = train_test_split(from_numpy(a), test_size=0.3, shuffle=False)
    tr = Xtr.gather().cpu().numpy().ravel()
    te = Xte.gather().cpu().numpy().ravel()
    np.testing.assert_array_equal(tr, np.arange(70))
    np.testing.assert_array_equal(te, np.arange(70, 100))
    kf = KFold(n_splits=4, shuffle=True, random_state=0)
    X = from_numpy(np.zeros((1000, 2), dtype=np.float32))
    seen = np.zeros(1000)
    for tr_m, te_m in kf.split(X):
        te_np = te_m.cpu().numpy()
        assert 150 < te_np.sum() < 350  # ~250 each in expectation
        seen += te_np
    np.testing.assert_array_equal(seen, 1.0)  # every row in exactly 1 fold


def test_label_encoder_categorical_fast_path():
    import pandas as pd
    s = pd.Series(pd.Categorical(["b", "a", "c", "a", "b"],
                                 categories=["a", "b", "c"]))
    le = LabelEncoder().fit(s)
    np.testing.assert_array_equal(le.classes_, ["a", "b", "c"])
    codes = le.transform(s)
    np.testing.assert_array_equal(codes, [1, 0, 2, 0, 1])
    back = le.inverse_transform(codes)
    assert list(back) == ["b", "a", "c", "a", "b"]
    # unseen label -> error
    s2 = pd.Series(pd.Categorical(["a", "z"], categories=["a", "z"]))
    with pytest.raises(ValueError, match="unseen"):
        le.transform(s2)
    # DistList of categorical partitions
    from mi355ml.core.bag import DistList
    dl = DistList([s[:3], s[3:]])
    le2 = LabelEncoder().fit(dl)
    np.testing.assert_array_equal(le2.classes_, ["a", "b", "c"])
    out = le2.transform(dl).gather()
    np.testing.assert_array_equal(np.concatenate(out), [1, 0, 2, 0, 1])
    # mismatched partition categories refuse (run Categorizer first)
    bad = DistList([s[:3], pd.Series(pd.Categorical(["x"]))])
    with pytest.raises(ValueError, match="mismatched"):
        LabelEncoder().fit(bad)


def test_polynomial_features_arbitrary_degree():
    # arbitrary degree + interaction_only vs sklearn
    rng = np.random.default_rng(21)
    a = rng.standard_normal((50, 3)).astype(np.float32)
    for degree, inter in [(4, False), (5, False), (3, True), (1, False),
                          (0, False)]:
        pf = PolynomialFeatures(degree=degree, interaction_only=inter)
        ours = pf.fit(from_numpy(a)).transform(from_numpy(a))
        sk = SkPoly(degree=degree, interaction_only=inter).fit_transform(
            a.astype(np.float64))
        got = ours.gather().cpu().numpy()
        assert got.shape == sk.shape, (degree, inter, got.shape, sk.shape)
        np.testing.assert_allclose(got, sk, rtol=2e-3, atol=2e-3)
    with pytest.raises(ValueError, match="refusing"):
        PolynomialFeatures(degree=9).fit(
            from_numpy(rng.standard_normal((10, 30)).astype(np.float32)))
