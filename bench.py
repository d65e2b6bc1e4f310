"""mi355ml flagship benchmark — BASELINE.json metric:
samples/sec (whole node), KMeans k=100 + LogReg fit on synthetic fp32
1e8x100 at 1/2/4/8 GPUs (STRONG scaling: the global problem is the named
1e8x100 shape at every N; per-GPU shard = 1e8/N rows).

One step = one LogisticRegression objective+gradient evaluation + coefficient
update (fused HIP kernel) PLUS one KMeans Lloyd iteration (fused
assign+partials HIP kernel) over the full shard, with the GLM [d+2] and
KMeans [k*d+k+1] statistics fused into ONE allreduce bucket per step
(VERDICT r1 #1a) — the steady-state hot loop of BASELINE configs 2 and 3
(SURVEY §3.2/§3.3).

Extra driver-parseable modes, same JSON contract:
  --config 2   LogisticRegression full lbfgs fit, 1e8x100 fp32
  --config 3   KMeans k=100 full fit (k-means|| init + Lloyd), 1e9x50 fp32
  --config 4   TruncatedSVD n_components=50, randomized, 1e8x1000 fp32
  --config 5   Hyperband over SGDClassifier.partial_fit, 81 configs
(3/4 honestly cap rows-per-GPU to fit HBM at small N and report the actual
shape in config.)

Launch (driver contract):
  python bench.py --gpus 1 --steps K --warmup W
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W
"""

from __future__ import annotations

import argparse
import json
import time

import torch

import mi355ml  # noqa: F401
from mi355ml.core import random as mlrand
from mi355ml.core.array import DistArray
from mi355ml.ops import glm_value_grad, kmeans_assign
from mi355ml.parallel import comm as comm_mod

METRIC = ("samples/sec (whole node) KMeans k=100 + LogReg fit, "
          "1e8x100 fp32, 1/2/4/8 GPU")


def build_data(rows_local: int, d: int, k: int, comm, dtype=torch.float32,
               rows_global: int | None = None,
               row_offset: int | None = None):
    """Synthetic blobs + logistic labels, generated per-shard on device
    (random-init equivalent; no dataset downloads — BASELINE.md rules)."""
    dev = comm.device
    mg = mlrand.generator(12345, shard_id=0xBEEF, device=dev)
    centers = torch.randn(k, d, generator=mg, device=dev, dtype=dtype) * 3.0
    w_true = torch.randn(d, generator=mg, device=dev, dtype=dtype) * 0.2
    sg = mlrand.generator(12345, shard_id=comm.rank, device=dev)
    lab = torch.randint(0, k, (rows_local,), generator=sg, device=dev)
    x = torch.randn(rows_local, d, generator=sg, device=dev, dtype=dtype)
    x += centers[lab]
    y = (torch.sigmoid(x @ w_true)
         > torch.rand(rows_local, generator=sg, device=dev)).to(dtype)
    total = rows_global or rows_local * comm.world_size
    off = row_offset if row_offset is not None else comm.rank * rows_local
    X = DistArray(x.contiguous(), total, off, comm)
    Y = DistArray(y.contiguous(), total, off, comm)
    return X, Y, centers.clone()


def one_step(X, Y, w, b, centers, lr, comm, buf):
    """Fused hot-loop step: GLM kernel + KMeans kernel back-to-back on the
    compute stream, then ONE allreduce of the combined [d+2 + k*d+k+1]
    statistics bucket (issued async on the comm side-stream)."""
    d = X.ncols
    k = centers.shape[0]
    loss, gw, gb = glm_value_grad(X.local, Y.local, w, b, "logistic")
    labels, sums, counts, inertia = kmeans_assign(X.local, centers)
    buf[0] = loss
    buf[1] = gb
    buf[2:2 + d] = gw
    o = 2 + d
    buf[o:o + k * d] = sums.reshape(-1)
    buf[o + k * d:o + k * d + k] = counts
    buf[-1] = inertia
    if comm.world_size > 1:
        comm.all_reduce_async(buf).wait()
    n = float(X.global_rows)
    w = (w.double() - lr * buf[2:2 + d] / n).float()
    b = b - lr * float(buf[1]) / n
    gcounts = buf[o + k * d:o + k * d + k].clamp(min=1.0)
    centers = (buf[o:o + k * d].reshape(k, d) / gcounts[:, None]).to(X.dtype)
    return w, b, centers, float(buf[0]), float(buf[-1])


def _sync(comm, on_gpu):
    comm.barrier()
    if on_gpu:
        torch.cuda.synchronize()


def _max_over_ranks(comm, elapsed, on_gpu):
    t = torch.tensor([elapsed], dtype=torch.float64,
                     device=comm.device if on_gpu else "cpu")
    if comm.world_size > 1:
        comm.all_reduce(t, op="max")
    return float(t[0])


def _counters():
    """Per-collective call/byte counters for the bench report
    (VERDICT r1 #9 / BASELINE.md observability promise)."""
    from mi355ml.diagnostics import COUNTERS
    return {k: v for k, v in COUNTERS.snapshot().items()
            if k.startswith("comm.")}


def emit(comm, *, metric, value, unit, args, ms_per_step, config,
         higher_is_better=True, scaling="strong", steps=None, dtype="fp32"):
    if comm.rank != 0:
        return
    config = dict(config)
    config["counters"] = _counters()
    out = {
        "metric": metric,
        "value": value,
        "unit": unit,
        "n_gpus": comm.world_size,
        "steps": steps if steps is not None else args.steps,
        "warmup": args.warmup,
        "ms_per_step": ms_per_step,
        "higher_is_better": higher_is_better,
        "scaling": scaling,
        "vs_baseline": None,
        "dtype": dtype,
        "data": "synthetic",
        "config": config,
    }
    print(json.dumps(out))


# ---------------------------------------------------------------------------
# headline (default): fused GLM+KMeans step over the named 1e8x100 shape
# ---------------------------------------------------------------------------

def bench_headline(args, comm):
    on_gpu = comm.device.type == "cuda"
    rows_global = args.rows or (100_000_000 if on_gpu else 100_000)
    base, rem = divmod(rows_global, comm.world_size)
    rows_local = base + (1 if comm.rank < rem else 0)
    row_offset = comm.rank * base + min(comm.rank, rem)
    d, k = args.features, args.clusters

    X, Y, centers = build_data(rows_local, d, k, comm,
                               rows_global=rows_global,
                               row_offset=row_offset)
    w = torch.zeros(d, dtype=torch.float32, device=comm.device)
    b = 0.0
    lr = 1.0
    buf = torch.empty(2 + d + k * d + k + 1, dtype=torch.float64,
                      device=X.device)

    for _ in range(args.warmup):
        w, b, centers, loss, inertia = one_step(X, Y, w, b, centers, lr,
                                                comm, buf)
    _sync(comm, on_gpu)
    t0 = time.perf_counter()
    for _ in range(args.steps):
        w, b, centers, loss, inertia = one_step(X, Y, w, b, centers, lr,
                                                comm, buf)
    _sync(comm, on_gpu)
    elapsed = _max_over_ranks(comm, time.perf_counter() - t0, on_gpu)

    # untimed single-step kernel probe (HIP events) for the report
    probe = {}
    if on_gpu:
        ev = [torch.cuda.Event(enable_timing=True) for _ in range(4)]
        ev[0].record()
        glm_value_grad(X.local, Y.local, w, b, "logistic")
        ev[1].record()
        kmeans_assign(X.local, centers)
        ev[2].record()
        if comm.world_size > 1:
            comm.all_reduce(buf)
        ev[3].record()
        torch.cuda.synchronize()
        from mi355ml.ops import kmeans as _km
        probe = {"glm_kernel_ms": ev[0].elapsed_time(ev[1]),
                 "kmeans_kernel_ms": ev[1].elapsed_time(ev[2]),
                 "allreduce_ms": ev[2].elapsed_time(ev[3]),
                 "kmeans_path": _km.last_path}

    emit(comm,
         metric=METRIC,
         value=rows_global * args.steps / elapsed,
         unit="samples/s", args=args,
         ms_per_step=elapsed / args.steps * 1000.0,
         config={
             "model": "LogisticRegression(lbfgs objective) + KMeans(k=100) "
                      "per-step Lloyd+grad, one fused allreduce bucket",
             "global_batch": rows_global,
             "rows_per_gpu": rows_local,
             "n_features": d,
             "n_clusters": k,
             "parallelism": f"dp{comm.world_size}",
             "final_loss": loss,
             "final_inertia": inertia,
             "probe_ms": probe,
         })


# ---------------------------------------------------------------------------
# --config 2: full LogisticRegression lbfgs fit on 1e8x100
# ---------------------------------------------------------------------------

def bench_config2(args, comm):
    from mi355ml.linear_model import LogisticRegression
    on_gpu = comm.device.type == "cuda"
    rows_global = args.rows or (100_000_000 if on_gpu else 50_000)
    rows_local = rows_global // comm.world_size
    d = 100
    X, Y, _ = build_data(rows_local, d, 16, comm, rows_global=rows_global)

    # warmup: short fit (kernel/launch caches)
    LogisticRegression(solver="lbfgs", max_iter=max(args.warmup, 1),
                       tol=0.0).fit(X, Y)
    _sync(comm, on_gpu)
    t0 = time.perf_counter()
    model = LogisticRegression(solver="lbfgs", max_iter=args.steps,
                               tol=0.0).fit(X, Y)
    _sync(comm, on_gpu)
    elapsed = _max_over_ranks(comm, time.perf_counter() - t0, on_gpu)
    nit = int(model.n_iter_)

    emit(comm,
         metric="samples*lbfgs_iters/sec LogisticRegression fit "
                "1e8x100 fp32 (BASELINE config 2)",
         value=rows_global * nit / elapsed,
         unit="samples*iters/s", args=args, steps=nit,
         ms_per_step=elapsed / max(nit, 1) * 1000.0,
         config={
             "model": "LogisticRegression(solver=lbfgs, tol=0)",
             "global_batch": rows_global,
             "rows_per_gpu": rows_local,
             "n_features": d,
             "parallelism": f"dp{comm.world_size}",
             "lbfgs_iterations": nit,
             "note": "one step = one lbfgs iteration (line-search obj "
                     "evals included in elapsed)",
         })


# ---------------------------------------------------------------------------
# --config 3: KMeans full fit (k-means|| init + Lloyd) on 1e9x50
# ---------------------------------------------------------------------------

def bench_config3(args, comm):
    from mi355ml.cluster import KMeans
    on_gpu = comm.device.type == "cuda"
    # full 1e9 rows needs >=2 GPUs (100 GB/GPU at N=2); cap honestly at N=1
    cap_per_gpu = 500_000_000 if on_gpu else 50_000
    rows_global = args.rows or min(1_000_000_000 if on_gpu else 100_000,
                                   cap_per_gpu * comm.world_size)
    rows_local = rows_global // comm.world_size
    d, k = 50, 100
    X, _, _ = build_data(rows_local, d, k, comm, rows_global=rows_global)

    # warmup on a small row slice: same kernels, untimed
    wrows = min(rows_local, 1_000_000)
    Xw = DistArray(X.local[:wrows], wrows * comm.world_size,
                   comm.rank * wrows, comm)
    KMeans(n_clusters=k, max_iter=max(args.warmup, 1), tol=0.0,
           random_state=0).fit(Xw)
    _sync(comm, on_gpu)
    t0 = time.perf_counter()
    model = KMeans(n_clusters=k, init="k-means||", max_iter=args.steps,
                   tol=0.0, random_state=0).fit(X)
    _sync(comm, on_gpu)
    elapsed = _max_over_ranks(comm, time.perf_counter() - t0, on_gpu)
    nit = int(model.n_iter_)

    emit(comm,
         metric="samples*lloyd_iters/sec KMeans k=100 full fit "
                "1e9x50 fp32 (BASELINE config 3)",
         value=rows_global * nit / elapsed,
         unit="samples*iters/s", args=args, steps=nit,
         ms_per_step=elapsed / max(nit, 1) * 1000.0,
         config={
             "model": "KMeans(k=100, init=k-means||, tol=0)",
             "global_batch": rows_global,
             "rows_per_gpu": rows_local,
             "n_features": d,
             "n_clusters": k,
             "parallelism": f"dp{comm.world_size}",
             "lloyd_iterations": nit,
             "inertia": float(model.inertia_),
             "note": "elapsed includes the k-means|| init passes "
                     "(part of the named config)",
         })


# ---------------------------------------------------------------------------
# --config 4: TruncatedSVD n_components=50 on 1e8x1000 (randomized)
# ---------------------------------------------------------------------------

def bench_config4(args, comm):
    from mi355ml.decomposition import TruncatedSVD
    on_gpu = comm.device.type == "cuda"
    # 1e8x1000 fp32 = 400 GB: needs >=4 GPUs; cap per-GPU honestly below
    cap_per_gpu = 30_000_000 if on_gpu else 20_000
    rows_global = args.rows or min(100_000_000 if on_gpu else 20_000,
                                   cap_per_gpu * comm.world_size)
    rows_local = rows_global // comm.world_size
    d = 1000
    dev = comm.device
    sg = mlrand.generator(777, shard_id=comm.rank, device=dev)
    x = torch.randn(rows_local, d, generator=sg, device=dev,
                    dtype=torch.float32)
    X = DistArray(x, rows_global, comm.rank * rows_local, comm)

    def one_fit():
        TruncatedSVD(n_components=50, algorithm="randomized",
                     n_iter=2, random_state=0).fit(X)

    for _ in range(max(args.warmup, 1)):
        one_fit()
    _sync(comm, on_gpu)
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_fit()
    _sync(comm, on_gpu)
    elapsed = _max_over_ranks(comm, time.perf_counter() - t0, on_gpu)

    emit(comm,
         metric="samples/sec TruncatedSVD k=50 randomized 1e8x1000 fp32 "
                "(BASELINE config 4)",
         value=rows_global * args.steps / elapsed,
         unit="samples/s", args=args,
         ms_per_step=elapsed / args.steps * 1000.0,
         config={
             "model": "TruncatedSVD(n_components=50, algorithm=randomized, "
                      "n_iter=2)",
             "global_batch": rows_global,
             "rows_per_gpu": rows_local,
             "n_features": d,
             "parallelism": f"dp{comm.world_size}",
             "note": "one step = one full fit (range finder + power "
                     "iterations + TSQR)",
         })


# ---------------------------------------------------------------------------
# --config 5: Hyperband over SGDClassifier.partial_fit, 81 configs
# ---------------------------------------------------------------------------

def bench_config5(args, comm):
    import numpy as np
    from sklearn.linear_model import SGDClassifier
    from mi355ml.model_selection import HyperbandSearchCV
    on_gpu = comm.device.type == "cuda"
    rows_local = 100_000 if on_gpu else 4_000
    rows_global = rows_local * comm.world_size
    d = 20
    X, Y, _ = build_data(rows_local, d, 4, comm, rows_global=rows_global)
    # binarize labels for the classifier
    yb = (Y.local > 0.5).to(torch.float32)
    Y = DistArray(yb, rows_global, comm.rank * rows_local, comm)

    params = {"alpha": np.logspace(-6, 0, 25).tolist(),
              "l1_ratio": [0.0, 0.15, 0.5, 0.85, 1.0]}

    def one_fit():
        hb = HyperbandSearchCV(
            SGDClassifier(loss="log_loss", penalty="elasticnet",
                          random_state=0),
            params, max_iter=81, random_state=0)
        hb.fit(X, Y)
        return hb

    for _ in range(max(args.warmup, 1)):
        hb = one_fit()
    _sync(comm, on_gpu)
    t0 = time.perf_counter()
    for _ in range(args.steps):
        hb = one_fit()
    _sync(comm, on_gpu)
    elapsed = _max_over_ranks(comm, time.perf_counter() - t0, on_gpu)
    pf_calls = int(hb.metadata_["partial_fit_calls"])
    train_rows = int(rows_local * 0.85)
    pf_samples = pf_calls * train_rows  # one slab == the owner's train shard

    emit(comm,
         metric="partial_fit samples/sec Hyperband 81 configs over "
                "SGDClassifier (BASELINE config 5)",
         value=pf_samples * args.steps / elapsed,
         unit="partial_fit samples/s", args=args,
         ms_per_step=elapsed / args.steps * 1000.0,
         scaling="weak",
         config={
             "model": "HyperbandSearchCV(SGDClassifier, max_iter=81)",
             "global_batch": rows_global,
             "rows_per_gpu": rows_local,
             "n_features": d,
             "n_models": int(hb.metadata_["n_models"]),
             "partial_fit_calls": pf_calls,
             "parallelism": f"trials@dp{comm.world_size}",
             "note": "one step = one full Hyperband fit; trials pinned "
                     "round-robin to ranks",
         })


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--config", type=str, default="headline",
                    choices=["headline", "2", "3", "4", "5"])
    ap.add_argument("--rows", type=int, default=None,
                    help="override GLOBAL row count")
    ap.add_argument("--rows-per-gpu", type=int, default=None,
                    help="legacy alias: global rows = rows_per_gpu * world")
    ap.add_argument("--features", type=int, default=100)
    ap.add_argument("--clusters", type=int, default=100)
    args = ap.parse_args()

    comm = comm_mod.init()
    if args.rows is None and args.rows_per_gpu is not None:
        args.rows = args.rows_per_gpu * comm.world_size

    {"headline": bench_headline, "2": bench_config2, "3": bench_config3,
     "4": bench_config4, "5": bench_config5}[args.config](args, comm)
    comm_mod.shutdown()


if __name__ == "__main__":
    main()
