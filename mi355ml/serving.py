"""Model serving: a batch-inference HTTP server for fitted estimators.

The reference has no serving story beyond joblib persistence; production
deployment needs one. `create_app(model)` builds a Starlette ASGI app
exposing /health, /info, /predict, /predict_proba and /transform over a
joblib-loadable mi355ml (or sklearn) estimator; inference runs on this
process's device through the estimator's blockwise predict path.
(Plain starlette, not FastAPI: the image's fastapi/starlette versions are
mutually incompatible.)

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


def create_app(model: Any, max_batch: int = 65536,
               token: Optional[str] = None):
    from starlette.applications import Starlette
    from starlette.responses import JSONResponse, PlainTextResponse
    from starlette.routing import Route

    from .diagnostics import COUNTERS, device_memory

    async def health(request):
        return JSONResponse({"status": "ok"})

    async def metrics(request):
        """Prometheus text exposition of the process counters + HBM usage."""
        lines = []
        for k, v in sorted(COUNTERS.snapshot().items()):
            name = "mi355ml_" + k.replace(".", "_").replace("-", "_")
            lines.append(f"{name} {v}")
        for k, v in device_memory().items():
            lines.append(f"mi355ml_device_{k} {v}")
        return PlainTextResponse("\n".join(lines) + "\n")

    async def info(request):
        return JSONResponse({
            "model": type(model).__name__,
            "n_features_in": getattr(model, "n_features_in_", None),
            "params": {k: repr(v) for k, v in model.get_params().items()}
            if hasattr(model, "get_params") else {},
        })

    def _method(name: str):
        async def endpoint(request):
            if token is not None:
                auth = request.headers.get("authorization", "")
                if auth != f"Bearer {token}":
                    return JSONResponse({"detail": "unauthorized"},
                                        status_code=401)
            if not hasattr(model, name):
                return JSONResponse({"detail": f"model has no {name}()"},
                                    status_code=400)
            try:
                payload = await request.json()
            except Exception:
                return JSONResponse({"detail": "invalid JSON body"},
                                    status_code=422)
            data = payload.get("data") if isinstance(payload, dict) else None
            try:
                x = np.asarray(data, dtype=np.float32)
                assert x.ndim == 2
            except Exception:
                return JSONResponse({"detail": "data must be a 2-D float "
                                               "array under key 'data'"},
                                    status_code=422)
            if x.shape[0] > max_batch:
                return JSONResponse(
                    {"detail": f"batch of {x.shape[0]} rows exceeds the "
                               f"server limit of {max_batch}"},
                    status_code=413)
            try:
                res = getattr(model, name)(x)
            except Exception as e:  # surface model errors as 400s
                COUNTERS.inc("serving.errors")
                return JSONResponse({"detail": str(e)}, status_code=400)
            COUNTERS.inc("serving.requests")
            COUNTERS.inc("serving.rows", float(x.shape[0]))
            return JSONResponse({"result": _to_plain(res)})

        return endpoint

    return Starlette(routes=[
        Route("/health", health, methods=["GET"]),
        Route("/metrics", metrics, methods=["GET"]),
        Route("/info", info, methods=["GET"]),
        Route("/predict", _method("predict"), methods=["POST"]),
        Route("/predict_proba", _method("predict_proba"), methods=["POST"]),
        Route("/transform", _method("transform"), methods=["POST"]),
    ])


def main(argv: Optional[list] = None):
    import argparse

    import joblib
    import uvicorn

    ap = argparse.ArgumentParser(prog="mi355ml-serve")
    ap.add_argument("model", help="joblib file with a fitted estimator")
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=8000)
    ap.add_argument("--max-batch", type=int, default=65536,
                    help="largest accepted request batch (rows)")
    ap.add_argument("--token", default=None,
                    help="require 'Authorization: Bearer <token>' on "
                         "inference endpoints")
    args = ap.parse_args(argv)
    model = joblib.load(args.model)
    uvicorn.run(create_app(model, max_batch=args.max_batch,
                           token=args.token),
                host=args.host, port=args.port)


if __name__ == "__main__":
    main()
