"""Timings for the round-1 late additions: tdigest percentiles, sparse
one-hot + sparse GLM, Hyperband over the native GPU estimator, and a
refreshed KMeans full-fit number after the incremental-init-labeling
rewrite."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))
import numpy as np
import torch

import mi355ml
from mi355ml import datasets


def timeit(name, fn):
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    out = fn()
    torch.cuda.synchronize()
    print(f"{name}: {time.perf_counter() - t0:.3f} s")
    return out


def main():
    # tdigest vs default percentile sketch at scale
    X, _ = datasets.make_regression(n_samples=50_000_000, n_features=100,
                                    random_state=0)
    from mi355ml.ops.percentile import col_percentiles
    qs = list(np.linspace(1, 99, 25))
    for _ in range(1):
        col_percentiles(X, qs)  # warm
    timeit("percentiles default 5e7x100 (25 qs)",
           lambda: col_percentiles(X, qs))
    timeit("percentiles tdigest 5e7x100 (25 qs)",
           lambda: col_percentiles(X, qs, method="tdigest"))
    del X
    torch.cuda.empty_cache()

    # sparse one-hot -> sparse logreg vs dense path
    rng = np.random.default_rng(0)
    a = rng.integers(0, 64, (20_000_000, 8)).astype(np.float32)
    y = (a[:, 0] + a[:, 1] > 62).astype(np.float32)
    from mi355ml.core.array import from_numpy
    from mi355ml.linear_model import LogisticRegression
    from mi355ml.preprocessing import OneHotEncoder
    Xd = from_numpy(a)
    yd = from_numpy(y)
    enc = OneHotEncoder(sparse_output=True).fit(Xd)
    S = timeit("one-hot sparse transform 2e7x8 -> 512 cols",
               lambda: enc.transform(Xd))
    print(f"  nnz={S.nnz}  sparse bytes={S.nnz*12/1e9:.2f} GB "
          f"vs dense {2e7*512*4/1e9:.1f} GB")
    lr = LogisticRegression(C=1.0, max_iter=40)
    timeit("LogReg lbfgs fit on SPARSE one-hot [2e7, 512]",
           lambda: lr.fit(S, yd))
    D = timeit("densify one-hot", lambda: S.todense())
    lrd = LogisticRegression(C=1.0, max_iter=40)
    timeit("LogReg lbfgs fit on DENSE one-hot [2e7, 512]",
           lambda: lrd.fit(D, yd))
    del S, D, Xd, yd
    torch.cuda.empty_cache()

    # Hyperband over the native GPU estimator
    Xs, ys = datasets.make_classification(n_samples=1_000_000,
                                          n_features=50, n_informative=10,
                                          random_state=3)
    from mi355ml.model_selection import HyperbandSearchCV
    with mi355ml.config.set({"array.block_rows": 100_000}):
        hb = timeit(
            "Hyperband(native GPU LogReg partial_fit) 1e6x50",
            lambda: HyperbandSearchCV(
                LogisticRegression(),
                {"C": [10.0 ** i for i in range(-4, 3)],
                 "eta0": [0.003, 0.01, 0.03, 0.1, 0.3, 1.0]},
                max_iter=27, random_state=0).fit(Xs, ys))
    print(f"  best={hb.best_score_:.4f} models={hb.metadata_['n_models']} "
          f"pf_calls={hb.metadata_['partial_fit_calls']}")
    del Xs, ys
    torch.cuda.empty_cache()

    # KMeans full fit at the estimator-table shape, post init rewrite
    Xk, _ = datasets.make_blobs(n_samples=12_500_000, n_features=50,
                                centers=100, random_state=2)
    from mi355ml.cluster import KMeans
    km = KMeans(n_clusters=100, random_state=0, max_iter=20, tol=1e-4)
    timeit("KMeans k=100 full fit 1.25e7x50", lambda: km.fit(Xk))
    print(f"  n_iter={km.n_iter_}")


if __name__ == "__main__":
    main()
