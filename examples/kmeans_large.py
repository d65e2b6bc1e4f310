"""KMeans k=100 on a large row-sharded matrix (BASELINE config-3 shape).

8 GPUs: python -m mi355ml.parallel.launch --gpus 8 examples/kmeans_large.py
(1e9 x 50 fp32 = 25 GB per GPU at 8 ranks.)
"""
import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.join(_os.path.dirname(
    _os.path.abspath(__file__)), ".."))
import time

from mi355ml import datasets
from mi355ml.cluster import KMeans
from mi355ml.parallel import comm

c = comm.init()
# 1e9 at 8 GPU ranks; small smoke size on CPU-only machines
per_rank = 125_000_000 if c.device.type == "cuda" else 100_000
n = per_rank * c.world_size
X, _ = datasets.make_blobs(n_samples=n, n_features=50, centers=100,
                           random_state=0)
t0 = time.perf_counter()
km = KMeans(n_clusters=100, random_state=0, max_iter=20,
            checkpoint_path="kmeans.ckpt", checkpoint_every=5).fit(X)
if c.rank == 0:
    dt = time.perf_counter() - t0
    print(f"fit {n:.2e}x50 in {dt:.1f}s  iters={km.n_iter_} "
          f"inertia={km.inertia_:.3e}")
comm.shutdown()
