"""Measure the BASELINE.md configs directly (per-GPU / 1-GPU forms).

Config 1: StandardScaler.fit_transform 1e5x20 (CPU plumbing check).
Config 2: LogisticRegression (lbfgs) full fit on 1e8x100 fp32 — 1 GPU,
          exactly the BASELINE config.
Config 3 (per-GPU shard): KMeans k=100 full fit (k-means|| init) on
          1.25e8x50 = 1e9/8.
"""
import os, sys, time
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch
import mi355ml
from mi355ml import datasets
from mi355ml.preprocessing import StandardScaler
from mi355ml.linear_model import LogisticRegression
from mi355ml.cluster import KMeans


def timed(name, fn):
    t0 = time.perf_counter()
    out = fn()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"{name}: {dt:.3f} s")
    return out, dt


# config 1 (CPU-ish plumbing): run on whatever device
X1, _ = datasets.make_regression(n_samples=100_000, n_features=20,
                                 random_state=0)
(_, t1) = timed("config1 StandardScaler.fit_transform 1e5x20",
                lambda: StandardScaler().fit_transform(X1))

if torch.cuda.is_available():
    # config 2: LogReg lbfgs on 1e8x100 (40 GB fp32), 1 GPU
    X2, y2 = datasets.make_classification(n_samples=100_000_000,
                                          n_features=100, n_informative=20,
                                          random_state=1)
    lr = LogisticRegression(C=1.0, solver="lbfgs", max_iter=100)
    (_, t2) = timed("config2 LogisticRegression(lbfgs) fit 1e8x100", 
                    lambda: lr.fit(X2, y2))
    print(f"  n_iter={lr.n_iter_}  samples/s(fit)={1e8/t2:.3e}  "
          f"acc={lr.score(X2, y2):.4f}")
    del X2, y2
    torch.cuda.empty_cache()

    # config 3 per-GPU shard: KMeans full fit
    X3, _ = datasets.make_blobs(n_samples=125_000_000, n_features=50,
                                centers=100, random_state=2)
    km = KMeans(n_clusters=100, random_state=0, max_iter=20, tol=1e-4)
    (_, t3) = timed("config3/8 KMeans k=100 fit 1.25e8x50 (kmeans|| init)",
                    lambda: km.fit(X3))
    print(f"  n_iter={km.n_iter_}  inertia={km.inertia_:.4e}  "
          f"samples/s(per Lloyd iter)={1.25e8 * km.n_iter_ / t3:.3e}")
