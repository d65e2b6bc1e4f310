"""Wide-X perf probe: col-sharded LogReg + TSVD at
~1e5 x 50k on 1 GPU, plus chunked-fused kmeans k=1024 vs composed.
Run on the GPU box; numbers go into profiles/README.md."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))
import torch  # noqa: E402

import mi355ml  # noqa: F401,E402
from mi355ml.core.colarray import ColDistArray  # noqa: E402
from mi355ml.parallel import comm as comm_mod  # noqa: E402
from mi355ml.linear_model import LogisticRegression  # noqa: E402
from mi355ml.decomposition import TruncatedSVD  # noqa: E402


def timeit(name, fn, reps=5, warm=2):
    for _ in range(warm):
        out = fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        out = fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / reps
    print(f"{name}: {dt * 1e3:.1f} ms")
    return dt, out


def main():
    comm = comm_mod.init()
    dev = comm.device
    n, d = 100_000, 50_000  # 20 GB f32
    g = torch.Generator(device=dev).manual_seed(0)
    print(f"== wide-X {n}x{d} f32 ({n * d * 4 / 2**30:.0f} GiB) ==")
    x = torch.randn(n, d, generator=g, device=dev)
    wtrue = torch.zeros(d, device=dev)
    wtrue[:64] = torch.randn(64, generator=g, device=dev)
    y = (torch.sigmoid(x @ wtrue)
         > torch.rand(n, generator=g, device=dev)).float()
    X = ColDistArray(x, d, 0, comm)

    # one objective evaluation = z-pass + grad-pass = 2 X reads (40 GB)
    from mi355ml.ops.glm import col_value_grad
    w0 = torch.zeros(d, device=dev)
    dt, _ = timeit("col_value_grad (2-pass, 40 GB traffic)",
                   lambda: col_value_grad(X, y, w0, 0.0, "logistic"))
    print(f"  effective {n * d * 4 * 2 / dt / 1e12:.2f} TB/s")

    t0 = time.perf_counter()
    m = LogisticRegression(solver="lbfgs", max_iter=20, tol=0.0).fit(X, y)
    torch.cuda.synchronize()
    fit_s = time.perf_counter() - t0
    print(f"LogReg lbfgs 20 iters on 1e5x50k: {fit_s:.2f} s "
          f"({fit_s / 20 * 1e3:.0f} ms/iter), acc="
          f"{m.score(X, y.cpu().numpy()):.3f}")

    t0 = time.perf_counter()
    t = TruncatedSVD(n_components=50, algorithm="randomized", n_iter=2,
                     random_state=0)
    t.fit(X)
    torch.cuda.synchronize()
    print(f"TSVD k=50 randomized on 1e5x50k: {time.perf_counter()-t0:.2f} s")

    del x, X, y
    torch.cuda.empty_cache()

    # chunked-fused kmeans k=1024 vs composed
    from mi355ml.ops import kmeans as km
    m_, d_, k_ = 4_000_000, 64, 1024
    x = torch.randn(m_, d_, generator=g, device=dev)
    c = torch.randn(k_, d_, generator=g, device=dev)
    dt, _ = timeit("closest_center chunked-fused k=1024",
                   lambda: km.closest_center(x, c))
    print(f"  path={km.last_path}  "
          f"{m_ * d_ * k_ * 2 / dt / 1e12:.1f} TF")

    def composed():
        cn = (c * c).sum(dim=1)
        block = max(1, min(m_, 8_388_608 // k_))
        labels = torch.empty(m_, dtype=torch.int32, device=x.device)
        dists = torch.empty(m_, dtype=x.dtype, device=x.device)
        for lo in range(0, m_, block):
            hi = min(m_, lo + block)
            xb = x[lo:hi]
            dd = cn[None, :] - 2.0 * (xb @ c.t())
            md, lab = dd.min(dim=1)
            labels[lo:hi] = lab.to(torch.int32)
            dists[lo:hi] = (md + (xb * xb).sum(dim=1)).clamp_(min=0)
        return labels, dists

    dt2, _ = timeit("closest_center composed rocBLAS+argmin k=1024",
                    composed)
    print(f"  chunked-fused speedup: {dt2 / dt:.2f}x")
    comm_mod.shutdown()


if __name__ == "__main__":
    main()
