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


class MicroBatcher:
    """Dynamic request batching: concurrent requests for
    the same method are coalesced into ONE model call — rows are
    concatenated up to max_batch (waiting up to max_wait_ms for
    stragglers), the model runs once in a thread-pool executor, and each
    caller gets its row slice back. One GPU pass per micro-batch instead
    of one per request."""

    def __init__(self, model: Any, max_batch: int = 65536,
                 max_wait_ms: float = 2.0):
        self.model = model
        self.max_batch = max_batch
        self.max_wait = max_wait_ms / 1e3
        self.queue = None  # created lazily on the serving event loop
        self._task = None
        self.calls = 0  # model invocations (observability/tests)

    def _ensure(self):
        import asyncio
        if self.queue is None:
            self.queue = asyncio.Queue()
            self._task = asyncio.get_event_loop().create_task(self._worker())

    async def submit(self, name: str, x: np.ndarray):
        import asyncio
        self._ensure()
        fut = asyncio.get_event_loop().create_future()
        await self.queue.put((name, x, fut))
        return await fut

    async def _worker(self):
        import asyncio
        loop = asyncio.get_event_loop()
        while True:
            name, x, fut = await self.queue.get()
            batch = [(x, fut)]
            rows = x.shape[0]
            other = []
            deadline = loop.time() + self.max_wait
            while rows < self.max_batch:
                timeout = deadline - loop.time()
                if timeout <= 0 and self.queue.empty():
                    break
                try:
                    item = await asyncio.wait_for(self.queue.get(),
                                                  max(timeout, 0.0))
                except asyncio.TimeoutError:
                    break
                if item[0] != name or rows + item[1].shape[0] > \
                        self.max_batch:
                    other.append(item)  # different method / would overflow
                    continue
                batch.append((item[1], item[2]))
                rows += item[1].shape[0]
            for item in other:
                self.queue.put_nowait(item)
            xs = (np.concatenate([b[0] for b in batch], axis=0)
                  if len(batch) > 1 else batch[0][0])

            def run():
                return _to_plain(getattr(self.model, name)(xs))

            try:
                self.calls += 1
                res = await loop.run_in_executor(None, run)
                lo = 0
                for xb, f in batch:
                    if not f.done():
                        f.set_result(res[lo:lo + xb.shape[0]])
                    lo += xb.shape[0]
            except Exception as e:
                for _, f in batch:
                    if not f.done():
                        f.set_exception(e)


def create_app(model: Any, max_batch: int = 65536,
               token: Optional[str] = None, batching: bool = False,
               max_wait_ms: float = 2.0):
    from starlette.applications import Starlette
    from starlette.responses import JSONResponse, PlainTextResponse
    from starlette.routing import Route

    from .diagnostics import COUNTERS, device_memory

    batcher = (MicroBatcher(model, max_batch, max_wait_ms)
               if batching else None)

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
                if batcher is not None:
                    plain = await batcher.submit(name, x)
                else:
                    plain = _to_plain(getattr(model, name)(x))
            except Exception as e:  # surface model errors as 400s
                COUNTERS.inc("serving.errors")
                return JSONResponse({"detail": str(e)}, status_code=400)
            COUNTERS.inc("serving.requests")
            COUNTERS.inc("serving.rows", float(x.shape[0]))
            return JSONResponse({"result": plain})

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
    import os

    ap = argparse.ArgumentParser(prog="mi355ml-serve")
    ap.add_argument("model", help="joblib file with a fitted estimator")
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=8000)
    ap.add_argument("--device", type=int, default=None,
                    help="pin this server process to one GPU (run one "
                         "replica per GPU behind a load balancer; ports "
                         "port+device by convention)")
    ap.add_argument("--max-batch", type=int, default=65536,
                    help="largest accepted request batch (rows)")
    ap.add_argument("--token", default=None,
                    help="require 'Authorization: Bearer <token>' on "
                         "inference endpoints")
    ap.add_argument("--batching", action="store_true",
                    help="coalesce concurrent requests into one model call")
    ap.add_argument("--max-wait-ms", type=float, default=2.0,
                    help="micro-batch straggler window (with --batching)")
    args = ap.parse_args(argv)
    if args.device is not None:
        # must happen before torch initializes HIP in joblib.load
        os.environ["HIP_VISIBLE_DEVICES"] = str(args.device)
        os.environ["CUDA_VISIBLE_DEVICES"] = str(args.device)
    import joblib
    import uvicorn
    model = joblib.load(args.model)
    port = args.port + (args.device or 0)
    uvicorn.run(create_app(model, max_batch=args.max_batch,
                           token=args.token, batching=args.batching,
                           max_wait_ms=args.max_wait_ms),
                host=args.host, port=port)


if __name__ == "__main__":
    main()
