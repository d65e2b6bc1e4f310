"""Ablation probe for the fused kmeans kernel (run on GPU box)."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch
import mi355ml  # noqa
from mi355ml.ops._ext import hip_ext

ext = hip_ext()
m, d, k = 4_000_000, 100, 100
g = torch.Generator(device="cuda").manual_seed(0)
x = torch.randn(m, d, generator=g, device="cuda")
c = torch.randn(k, d, generator=g, device="cuda")


def timeit(name, fn, reps=10):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(reps):
        fn()
    e.record()
    torch.cuda.synchronize()
    ms = s.elapsed_time(e) / reps
    print(f"{name:28s} {ms:8.3f} ms   ({m*d*4/ms*1e3/1e12:.2f} TB/s eff, "
          f"{2*m*d*((k+15)//16*16)/ms*1e3/1e12:.1f} TF if GEMM)")


timeit("closest_center (no update)", lambda: ext.closest_center(x, c))
timeit("kmeans_assign (update)", lambda: ext.kmeans_assign(x, c))
a = torch.rand(d, device="cuda") + 1
b = torch.rand(d, device="cuda")
out = torch.empty_like(x)
timeit("affine (HBM r+w reference)", lambda: ext.affine_transform(x, a, b, out))
timeit("rocblas x@c.T", lambda: x @ c.t())
la = torch.empty(m, dtype=torch.int32, device="cuda")
timeit("glm_value_grad", lambda: ext.glm_value_grad(
    x, x[:, 0].contiguous(), c[0].contiguous(), 0.1, 0))

print("--- phase ablation (bit0 stage, bit1 compute, bit2 update) ---")
for ph, name in [(0, "loop shell only"), (1, "stage"),
                 (3, "stage+mfma (epilogue dead)"),
                 (11, "stage+mfma+epilogue"),
                 (13, "stage+epi?+update"), (15, "all")]:
    timeit(f"phases={ph} ({name})",
           lambda ph=ph: ext.kmeans_phase_probe(x, c, ph))

print("--- glm phase ablation (bit0 stage, bit1 z, bit2 grad) ---")
yv = x[:, 0].contiguous()
wv = c[0].contiguous()
for ph, name in [(0, "shell"), (1, "stage"), (3, "stage+z"),
                 (5, "stage+grad"), (7, "all")]:
    timeit(f"glm phases={ph} ({name})",
           lambda ph=ph: ext.glm_phase_probe(x, yv, wv, ph))
