"""sklearn API contract across the estimator surface: clone /
get_params/set_params roundtrip, joblib persistence of fitted models
(SURVEY §2.1 model-persistence contract)."""

import joblib
import numpy as np
import pytest
from sklearn.base import clone

from mi355ml.core.array import from_numpy
from mi355ml.cluster import KMeans, SpectralClustering
from mi355ml.decomposition import PCA, IncrementalPCA, TruncatedSVD
from mi355ml.ensemble import BlockwiseVotingClassifier
from mi355ml.impute import SimpleImputer
from mi355ml.linear_model import (LinearRegression, LogisticRegression,
                                  PoissonRegression)
from mi355ml.naive_bayes import GaussianNB
from mi355ml.preprocessing import (BlockTransformer, LabelEncoder,
                                   MinMaxScaler, Normalizer, OneHotEncoder,
                                   OrdinalEncoder, PolynomialFeatures,
                                   QuantileTransformer, RobustScaler,
                                   StandardScaler)
from mi355ml.model_selection import (GridSearchCV, HyperbandSearchCV,
                                     IncrementalSearchCV,
                                     RandomizedSearchCV,
                                     SuccessiveHalvingSearchCV)
from mi355ml.wrappers import Incremental, ParallelPostFit

ALL_ESTIMATORS = [
    KMeans(), SpectralClustering(), PCA(), TruncatedSVD(), IncrementalPCA(),
    SimpleImputer(), LinearRegression(), LogisticRegression(),
    PoissonRegression(), GaussianNB(), MinMaxScaler(), Normalizer(),
    OneHotEncoder(), OneHotEncoder(sparse_output=True, drop="first"),
    OrdinalEncoder(), PolynomialFeatures(),
    QuantileTransformer(), QuantileTransformer(method="tdigest"),
    RobustScaler(), RobustScaler(method="tdigest"), StandardScaler(),
    LabelEncoder(),
    GridSearchCV(LogisticRegression(), {"C": [1.0]}),
    RandomizedSearchCV(LogisticRegression(), {"C": [1.0, 2.0]}, n_iter=2),
    HyperbandSearchCV(LogisticRegression(), {"C": [1.0, 2.0]}, max_iter=9),
    SuccessiveHalvingSearchCV(LogisticRegression(), {"C": [1.0, 2.0]}),
    IncrementalSearchCV(LogisticRegression(), {"C": [1.0, 2.0]}),
    Incremental(LogisticRegression()), ParallelPostFit(LogisticRegression()),
]


@pytest.mark.parametrize("est", ALL_ESTIMATORS,
                         ids=lambda e: type(e).__name__)
def test_clone_and_params_roundtrip(est):
    c = clone(est)
    params = est.get_params()
    c.set_params(**params)
    assert c.get_params() == params


def test_joblib_roundtrip_fitted_estimators(tmp_path):
    rng = np.random.default_rng(0)
    X = rng.standard_normal((500, 6)).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.float32)
    fitted = [
        StandardScaler().fit(from_numpy(X)),
        LogisticRegression(max_iter=20).fit(from_numpy(X), from_numpy(y)),
        PCA(n_components=3).fit(from_numpy(X)),
        KMeans(n_clusters=3, random_state=0, max_iter=10).fit(from_numpy(X)),
    ]
    for est in fitted:
        p = tmp_path / f"{type(est).__name__}.joblib"
        joblib.dump(est, p)
        est2 = joblib.load(p)
        if hasattr(est, "transform"):
            a = est.transform(from_numpy(X)).gather().cpu().numpy()
            b = est2.transform(from_numpy(X)).gather().cpu().numpy()
        else:
            a = est.predict(from_numpy(X)).gather().cpu().numpy()
            b = est2.predict(from_numpy(X)).gather().cpu().numpy()
        np.testing.assert_allclose(a, b, rtol=1e-6, atol=1e-6)


def test_joblib_backend_registers():
    import mi355ml.joblib_backend as jb
    jb.register()
    import joblib
    from sklearn.linear_model import SGDClassifier
    rng = np.random.default_rng(1)
    X = rng.standard_normal((200, 4))
    y = (X[:, 0] > 0).astype(int)
    with joblib.parallel_backend("mi355ml"):
        res = joblib.Parallel(n_jobs=2)(
            joblib.delayed(lambda s: SGDClassifier(random_state=s)
                           .fit(X, y).score(X, y))(s) for s in range(4))
    assert len(res) == 4
