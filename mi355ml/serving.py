"""Model serving: a batch-inference HTTP server for fitted estimators.

The reference has no serving story beyond joblib persistence; production
deployment needs one. `create_app(model)` builds a FastAPI app exposing
/health, /info, /predict and /predict_proba over a joblib-loadable
mi355ml (or sklearn) estimator; inference runs on this process's device
through the estimator's blockwise predict path.

    python -m mi355ml.serving model.joblib --host 0.0.0.0 --port 8000
"""

from __future__ import annotations

from typing import Any, List, Optional

import numpy as np

from .core.array import DistArray
from .core.bag import DistList


def _to_plain(res) -> List:
    if isinstance(res, DistArray):
        return res.gather().cpu().numpy().tolist()
    if isinstance(res, DistList):
        out = []
        for part in res.gather():
            arr = part.toarray() if hasattr(part, "toarray") else np.asarray(part)
            out.extend(np.asarray(arr).tolist())
        return out
    return np.asarray(res).tolist()


def create_app(model: Any):
    from fastapi import FastAPI, HTTPException
    from pydantic import BaseModel

    class PredictRequest(BaseModel):
        data: List[List[float]]

    app = FastAPI(title="mi355ml model server")
    app.state.model = model

    @app.get("/health")
    def health():
        return {"status": "ok"}

    @app.get("/info")
    def info():
        m = app.state.model
        return {
            "model": type(m).__name__,
            "n_features_in": getattr(m, "n_features_in_", None),
            "params": {k: repr(v) for k, v in m.get_params().items()}
            if hasattr(m, "get_params") else {},
        }

    def _run(name: str, req: PredictRequest):
        m = app.state.model
        if not hasattr(m, name):
            raise HTTPException(status_code=400,
                                detail=f"model has no {name}()")
        x = np.asarray(req.data, dtype=np.float32)
        if x.ndim != 2:
            raise HTTPException(status_code=422, detail="data must be 2-D")
        try:
            res = getattr(m, name)(x)
        except Exception as e:  # surface model errors as 400s
            raise HTTPException(status_code=400, detail=str(e))
        return {"result": _to_plain(res)}

    @app.post("/predict")
    def predict(req: PredictRequest):
        return _run("predict", req)

    @app.post("/predict_proba")
    def predict_proba(req: PredictRequest):
        return _run("predict_proba", req)

    @app.post("/transform")
    def transform(req: PredictRequest):
        return _run("transform", req)

    return app


def main(argv: Optional[list] = None):
    import argparse

    import joblib
    import uvicorn

    ap = argparse.ArgumentParser(prog="mi355ml-serve")
    ap.add_argument("model", help="joblib file with a fitted estimator")
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=8000)
    args = ap.parse_args(argv)
    model = joblib.load(args.model)
    uvicorn.run(create_app(model), host=args.host, port=args.port)


if __name__ == "__main__":
    main()
