"""Scalers vs sklearn oracle (exact for scalers)."""

import numpy as np
from sklearn.preprocessing import MinMaxScaler as SkMinMax
from sklearn.preprocessing import StandardScaler as SkStandard

from mi355ml.core.array import from_numpy
from mi355ml.preprocessing import MinMaxScaler, StandardScaler


def _data(n=2000, f=6, seed=0):
    rng = np.random.default_rng(seed)
    return (rng.standard_normal((n, f)) * rng.uniform(0.5, 5, f)
            + rng.uniform(-3, 3, f)).astype(np.float32)


def test_standard_scaler_matches_sklearn():
    a = _data()
    ours = StandardScaler().fit(from_numpy(a))
    ref = SkStandard().fit(a)
    np.testing.assert_allclose(ours.mean_.cpu().numpy(), ref.mean_, rtol=1e-6,
                               atol=1e-6)
    np.testing.assert_allclose(ours.scale_.cpu().numpy(), ref.scale_, rtol=1e-5)
    got = ours.transform(from_numpy(a)).gather().cpu().numpy()
    np.testing.assert_allclose(got, ref.transform(a), rtol=1e-4, atol=1e-4)
    back = ours.inverse_transform(from_numpy(got)).gather().cpu().numpy()
    np.testing.assert_allclose(back, a, rtol=1e-4, atol=1e-4)


def test_standard_scaler_zero_variance_column():
    a = _data()
    a[:, 2] = 7.0
    ours = StandardScaler().fit(from_numpy(a))
    got = ours.transform(from_numpy(a)).gather().cpu().numpy()
    assert np.allclose(got[:, 2], 0.0, atol=1e-5)


def test_minmax_scaler_matches_sklearn():
    a = _data(seed=1)
    ours = MinMaxScaler().fit(from_numpy(a))
    ref = SkMinMax().fit(a)
    np.testing.assert_allclose(ours.data_min_.cpu().numpy(), ref.data_min_,
                               rtol=1e-6)
    np.testing.assert_allclose(ours.data_max_.cpu().numpy(), ref.data_max_,
                               rtol=1e-6)
    got = ours.fit_transform(from_numpy(a)).gather().cpu().numpy()
    np.testing.assert_allclose(got, ref.transform(a), rtol=1e-4, atol=1e-5)


def _scaler_world2(rank, world, a_bytes, shape):
    a = np.frombuffer(a_bytes, dtype=np.float32).reshape(shape).copy()
    ref = SkStandard().fit(a)
    ours = StandardScaler().fit(from_numpy(a))
    np.testing.assert_allclose(ours.mean_.cpu().numpy(), ref.mean_, rtol=1e-6,
                               atol=1e-6)
    got = ours.transform(from_numpy(a)).gather().cpu().numpy()
    np.testing.assert_allclose(got, ref.transform(a), rtol=1e-4, atol=1e-4)


def test_standard_scaler_world2(mp_runner):
    a = _data(777, 4, seed=5)
    mp_runner(_scaler_world2, world=2, args=(a.tobytes(), a.shape))
