"""Golden-path integration: one flow through preprocessing -> PCA ->
GridSearchCV(LogReg pipeline) -> persistence -> serving, all on
row-sharded data (the end-to-end interop a dask-ml user would run)."""

import numpy as np
import pytest

from mi355ml.core.array import from_numpy


def test_end_to_end_workflow(tmp_path):
    import joblib
    from sklearn.pipeline import Pipeline
    from starlette.testclient import TestClient

    from mi355ml.decomposition import PCA
    from mi355ml.linear_model import LogisticRegression
    from mi355ml.model_selection import GridSearchCV, train_test_split
    from mi355ml.preprocessing import StandardScaler
    from mi355ml.serving import create_app

    rng = np.random.default_rng(42)
    n, d = 3000, 20
    X = rng.standard_normal((n, d)).astype(np.float32)
    X[:, :5] *= 8.0  # unscaled columns
    w = np.zeros(d); w[:8] = rng.standard_normal(8)
    y = (1 / (1 + np.exp(-((X / X.std(0)) @ w))) >
         rng.uniform(size=n)).astype(np.float32)

    Xtr, Xte, ytr, yte = train_test_split(from_numpy(X), from_numpy(y),
                                          test_size=0.25, random_state=0)
    pipe = Pipeline([
        ("scale", StandardScaler()),
        ("pca", PCA(n_components=10, svd_solver="randomized",
                    random_state=0)),
        ("clf", LogisticRegression(solver="lbfgs", max_iter=60)),
    ])
    gs = GridSearchCV(pipe, {"clf__C": [0.1, 1.0, 10.0]}, cv=3,
                      return_train_score=True)
    gs.fit(Xtr, ytr)
    assert gs.best_score_ > 0.65
    acc = gs.score(Xte, yte)
    assert acc > 0.65

    # persistence roundtrip of the refit pipeline
    path = tmp_path / "model.joblib"
    joblib.dump(gs.best_estimator_, path)
    loaded = joblib.load(path)
    p1 = loaded.predict(from_numpy(X[:64])).gather().cpu().numpy()
    p0 = gs.best_estimator_.predict(
        from_numpy(X[:64])).gather().cpu().numpy()
    np.testing.assert_array_equal(p0, p1)

    # serve the loaded model and predict over HTTP (micro-batching on)
    app = create_app(loaded, batching=True, max_wait_ms=1.0)
    with TestClient(app) as client:
        r = client.post("/predict", json={"data": X[:8].tolist()})
        assert r.status_code == 200
        np.testing.assert_array_equal(np.asarray(r.json()["result"]),
                                      p0[:8])
        assert client.get("/metrics").status_code == 200


@pytest.mark.gpu
def test_end_to_end_workflow_gpu(tmp_path):
    # same flow with device-resident shards (from_numpy lands on cuda:0)
    test_end_to_end_workflow(tmp_path)
