"""BASELINE config-3 per-GPU measurement: KMeans k=100 Lloyd on d=50 at the
8-GPU config's per-GPU shard size (1e9/8 = 1.25e8 rows x 50 = 25 GB fp32)."""
import os, sys, time
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch
import mi355ml  # noqa
from mi355ml.ops import kmeans_assign

m, d, k = 125_000_000, 50, 100
g = torch.Generator(device="cuda").manual_seed(0)
x = torch.randn(m, d, generator=g, device="cuda")
c = torch.randn(k, d, generator=g, device="cuda") * 2
for _ in range(2):
    kmeans_assign(x, c)
torch.cuda.synchronize()
t0 = time.perf_counter()
iters = 5
for _ in range(iters):
    labels, sums, cnt, inertia = kmeans_assign(x, c)
    c = (sums / cnt.clamp(min=1)[:, None]).float()
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / iters
print(f"Lloyd iter (1.25e8 x 50, k=100): {dt*1e3:.1f} ms -> "
      f"{m/dt:.3e} samples/s/GPU "
      f"({2*m*d*((k+15)//16*16)/dt/1e12:.1f} TF on the distance GEMM)")
