"""mi355ml flagship benchmark — BASELINE.json metric:
samples/sec (whole node), KMeans k=100 + LogReg fit on synthetic fp32
1e8x100 (weak scaling: 1.25e7 rows per GPU -> 1e8 rows at 8 GPUs).

One step = one LogisticRegression objective+gradient evaluation + coefficient
update (fused HIP kernel + one fused allreduce) PLUS one KMeans Lloyd
iteration (fused assign+partials HIP kernel + one fused allreduce + centroid
update) over the full shard — the steady-state hot loop of BASELINE configs
2 and 3.

Launch (driver contract):
  python bench.py --gpus 1 --steps K --warmup W
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W
"""

from __future__ import annotations

import argparse
import json
import os
import time

import torch

import mi355ml  # noqa: F401
from mi355ml.core import random as mlrand
from mi355ml.core.array import DistArray
from mi355ml.ops import glm_value_grad, kmeans_assign
from mi355ml.parallel import comm as comm_mod


def build_data(rows_local: int, d: int, k: int, comm, dtype=torch.float32):
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
    total = rows_local * comm.world_size
    X = DistArray(x.contiguous(), total, comm.rank * rows_local, comm)
    Y = DistArray(y.contiguous(), total, comm.rank * rows_local, comm)
    return X, Y, centers.clone()


def one_step(X, Y, w, b, centers, lr, comm):
    d = X.ncols
    k = centers.shape[0]
    # --- LogReg objective + gradient (fused kernel) + update ---
    loss, gw, gb = glm_value_grad(X.local, Y.local, w, b, "logistic")
    buf = torch.empty(2 + d, dtype=torch.float64, device=gw.device)
    buf[0] = loss
    buf[1] = gb
    buf[2:] = gw
    handle = comm.all_reduce_async(buf) if comm.world_size > 1 else None
    # --- KMeans Lloyd iteration (fused kernel) overlapped with the allreduce
    labels, sums, counts, inertia = kmeans_assign(X.local, centers)
    kbuf = torch.empty(k * d + k + 1, dtype=torch.float64, device=gw.device)
    kbuf[:k * d] = sums.reshape(-1)
    kbuf[k * d:k * d + k] = counts
    kbuf[-1] = inertia
    if comm.world_size > 1:
        comm.all_reduce(kbuf)
        handle.wait()
    n = float(X.global_rows)
    w = (w.double() - lr * buf[2:] / n).float()
    b = b - lr * float(buf[1]) / n
    gcounts = kbuf[k * d:k * d + k].clamp(min=1.0)
    centers = (kbuf[:k * d].reshape(k, d) / gcounts[:, None]).to(X.dtype)
    return w, b, centers, float(buf[0]), float(kbuf[-1])


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--rows-per-gpu", type=int, default=None)
    ap.add_argument("--features", type=int, default=100)
    ap.add_argument("--clusters", type=int, default=100)
    args = ap.parse_args()

    comm = comm_mod.init()
    on_gpu = comm.device.type == "cuda"
    rows_local = args.rows_per_gpu or (12_500_000 if on_gpu else 100_000)
    d, k = args.features, args.clusters

    X, Y, centers = build_data(rows_local, d, k, comm)
    w = torch.zeros(d, dtype=torch.float32, device=comm.device)
    b = 0.0
    lr = 1.0

    def sync():
        comm.barrier()
        if on_gpu:
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        w, b, centers, loss, inertia = one_step(X, Y, w, b, centers, lr, comm)
    sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        w, b, centers, loss, inertia = one_step(X, Y, w, b, centers, lr, comm)
    sync()
    elapsed = time.perf_counter() - t0

    # max over ranks
    t = torch.tensor([elapsed], dtype=torch.float64,
                     device=comm.device if on_gpu else "cpu")
    if comm.world_size > 1:
        comm.all_reduce(t, op="max")
    elapsed = float(t[0])

    total_rows = rows_local * comm.world_size
    ms_per_step = elapsed / args.steps * 1000.0
    samples_per_sec = total_rows * args.steps / elapsed

    if comm.rank == 0:
        out = {
            "metric": "samples/sec (whole node) KMeans k=100 + LogReg fit, "
                      "1e8x100 fp32, 1/2/4/8 GPU",
            "value": samples_per_sec,
            "unit": "samples/s",
            "n_gpus": comm.world_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic",
            "config": {
                "model": "LogisticRegression(lbfgs objective) + KMeans(k=100) "
                         "per-step Lloyd+grad",
                "global_batch": total_rows,
                "rows_per_gpu": rows_local,
                "n_features": d,
                "n_clusters": k,
                "parallelism": f"dp{comm.world_size}",
                "final_loss": loss,
                "final_inertia": inertia,
            },
        }
        print(json.dumps(out))
    comm_mod.shutdown()


if __name__ == "__main__":
    main()
