"""A/B the wave-specialized split kmeans kernel vs the in-line one:
bitwise equality + timing at the bench shape."""
import os, sys, time
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch
import mi355ml  # noqa
from mi355ml.ops._ext import hip_ext

ext = hip_ext()
g = torch.Generator(device="cuda").manual_seed(0)
for m, d, k in [(4_000_000, 100, 100), (4_000_000, 50, 100),
                (1_000_000, 37, 73), (123_457, 100, 100)]:
    x = torch.randn(m, d, generator=g, device="cuda")
    c = torch.randn(k, d, generator=g, device="cuda")
    la, sa, ca, ia = ext.kmeans_assign(x, c)
    lb, sb, cb, ib = ext.kmeans_assign_split(x, c)
    ok_l = bool((la == lb).all())
    # d<=64 routes to the 128-row-tile kernel whose block->row assignment
    # differs from the inline kernel -> f32 bin partials differ in low
    # bits (both are run-to-run deterministic); compare with tolerance
    if (d + 3) & ~3 <= 64 and k <= 128:
        ok_s = bool(torch.allclose(sa, sb, rtol=1e-6, atol=1e-2))
    else:
        ok_s = bool((sa == sb).all())
    ok_c = bool((ca == cb).all())
    # inertia: f64 reduction-tree order differs across thread counts AND
    # the per-lane f32 sx2 accumulation order differs between the 8-row
    # chain batches and the inline kernel's 4-term expressions -> agree
    # to ~1e-6 rel (real accumulation bugs show up at %-level)
    ok_i = abs(float(ia) - float(ib)) <= 1e-6 * max(abs(float(ia)), 1.0)
    print(f"m={m} d={d} k={k}: labels={ok_l} sums={ok_s} cnt={ok_c} inertia={ok_i}")
    assert ok_l and ok_s and ok_c and ok_i

    def t(fn, reps=10):
        for _ in range(3): fn()
        torch.cuda.synchronize(); t0 = time.perf_counter()
        for _ in range(reps): fn()
        torch.cuda.synchronize(); return (time.perf_counter()-t0)/reps*1e3
    ta = t(lambda: ext.kmeans_assign(x, c))
    tb = t(lambda: ext.kmeans_assign_split(x, c))
    print(f"  inline {ta:.3f} ms  split {tb:.3f} ms  speedup {ta/tb:.2f}x")
    del x, c
    torch.cuda.empty_cache()
print("ALL OK")
