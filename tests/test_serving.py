"""Model server: FastAPI app over a fitted estimator (TestClient, no net)."""

import numpy as np
from starlette.testclient import TestClient

from mi355ml.core.array import from_numpy
from mi355ml.linear_model import LogisticRegression
from mi355ml.serving import create_app


def test_serving_roundtrip():
    rng = np.random.default_rng(0)
    X = rng.standard_normal((500, 4)).astype(np.float32)
    y = (X[:, 0] + X[:, 1] > 0).astype(np.float32)
    lr = LogisticRegression(C=10.0, max_iter=50).fit(from_numpy(X),
                                                     from_numpy(y))
    client = TestClient(create_app(lr))
    assert client.get("/health").json() == {"status": "ok"}
    info = client.get("/info").json()
    assert info["model"] == "LogisticRegression"
    assert info["n_features_in"] == 4
    r = client.post("/predict", json={"data": X[:10].tolist()})
    assert r.status_code == 200
    pred = np.asarray(r.json()["result"])
    assert pred.shape == (10,)
    r = client.post("/predict_proba", json={"data": X[:5].tolist()})
    proba = np.asarray(r.json()["result"])
    assert proba.shape == (5, 2)
    np.testing.assert_allclose(proba.sum(axis=1), 1.0, rtol=1e-5)
    # malformed input -> 4xx, not a crash
    r = client.post("/predict", json={"data": "nope"})
    assert r.status_code == 422
    # metrics: request/row counters visible in prometheus text format
    m = client.get("/metrics")
    assert m.status_code == 200
    assert "mi355ml_serving_requests" in m.text
    assert "mi355ml_serving_rows" in m.text


def test_serving_batch_limit():
    rng = np.random.default_rng(1)
    X = rng.standard_normal((200, 3)).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.float32)
    lr = LogisticRegression(max_iter=20).fit(from_numpy(X), from_numpy(y))
    client = TestClient(create_app(lr, max_batch=50))
    r = client.post("/predict", json={"data": X[:51].tolist()})
    assert r.status_code == 413
    r = client.post("/predict", json={"data": X[:50].tolist()})
    assert r.status_code == 200


def test_serving_auth():
    rng = np.random.default_rng(2)
    X = rng.standard_normal((100, 3)).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.float32)
    lr = LogisticRegression(max_iter=10).fit(from_numpy(X), from_numpy(y))
    client = TestClient(create_app(lr, token="s3cret"))
    r = client.post("/predict", json={"data": X[:5].tolist()})
    assert r.status_code == 401
    r = client.post("/predict", json={"data": X[:5].tolist()},
                    headers={"Authorization": "Bearer s3cret"})
    assert r.status_code == 200
    # health stays open
    assert client.get("/health").status_code == 200


def test_micro_batcher_coalesces_concurrent_requests():
    """VERDICT r1 #10 serving batching: N concurrent submits for the same
    method produce FEWER model calls than requests, and every caller gets
    exactly its rows back."""
    import asyncio
    from mi355ml.serving import MicroBatcher

    class SlowModel:
        def __init__(self):
            self.calls = 0

        def predict(self, x):
            self.calls += 1
            import time
            time.sleep(0.05)  # force queue build-up while the first runs
            return np.asarray(x).sum(axis=1)

    model = SlowModel()
    mb = MicroBatcher(model, max_batch=1000, max_wait_ms=20.0)

    async def run():
        xs = [np.full((3, 2), float(i), dtype=np.float32) for i in range(6)]
        results = await asyncio.gather(
            *[mb.submit("predict", x) for x in xs])
        return xs, results

    xs, results = asyncio.new_event_loop().run_until_complete(run())
    for x, r in zip(xs, results):
        np.testing.assert_allclose(np.asarray(r), x.sum(axis=1))
    assert model.calls < 6, f"no batching happened ({model.calls} calls)"
    assert mb.calls == model.calls


def test_serving_with_batching_endpoint():
    from starlette.testclient import TestClient
    from mi355ml.serving import create_app

    class M:
        def predict(self, x):
            return np.asarray(x)[:, 0] * 2

    app = create_app(M(), batching=True, max_wait_ms=1.0)
    with TestClient(app) as client:
        r = client.post("/predict", json={"data": [[1.0, 2.0], [3.0, 4.0]]})
        assert r.status_code == 200
        assert r.json()["result"] == [2.0, 6.0]
