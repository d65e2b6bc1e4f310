"""Estimator-level GPU benchmark table (1 GPU) — evidence for profiles/.

Times the headline estimator fits on synthetic device data at per-GPU
BASELINE-ish scales.
"""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))
import torch  # noqa: E402

import mi355ml  # noqa: E402,F401
from mi355ml import datasets  # noqa: E402


def timeit(name, fn, reps=2, sync=True):
    """Run reps times, report the BEST (steady-state; first call pays
    hipMalloc pool growth and kernel-launch caches)."""
    best = None
    out = None
    for _ in range(max(reps, 1)):
        t0 = time.perf_counter()
        out = fn()
        if sync and torch.cuda.is_available():
            torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        best = dt if best is None else min(best, dt)
    print(f"{name:58s} {best:8.2f} s")
    return out


def main():
    rows = 12_500_000 if torch.cuda.is_available() else 100_000

    X, y = datasets.make_classification(n_samples=rows, n_features=100,
                                        n_informative=20, random_state=0)
    from mi355ml.linear_model import LogisticRegression
    lr = timeit(f"LogisticRegression(lbfgs) fit {rows:.0e}x100",
                lambda: LogisticRegression(C=1.0, max_iter=100).fit(X, y))
    print(f"    n_iter={lr.n_iter_}")

    from mi355ml.preprocessing import StandardScaler, RobustScaler
    timeit(f"StandardScaler fit_transform {rows:.0e}x100",
           lambda: StandardScaler().fit_transform(X))
    timeit(f"RobustScaler fit {rows:.0e}x100",
           lambda: RobustScaler().fit(X))

    Xb, _ = datasets.make_blobs(n_samples=rows, n_features=50, centers=100,
                                random_state=1)
    from mi355ml.cluster import KMeans
    km = timeit(f"KMeans k=100 fit (kmeans|| init) {rows:.0e}x50",
                lambda: KMeans(n_clusters=100, random_state=0,
                               max_iter=20).fit(Xb))
    print(f"    n_iter={km.n_iter_} inertia={km.inertia_:.3e}")

    del Xb
    Xw, _ = datasets.make_regression(
        n_samples=rows // 2, n_features=1000, random_state=2)
    from mi355ml.decomposition import TruncatedSVD, PCA
    timeit(f"TruncatedSVD k=50 randomized {rows//2:.0e}x1000",
           lambda: TruncatedSVD(n_components=50,
                                algorithm="randomized").fit(Xw))
    timeit(f"PCA k=50 tsqr {rows//2:.0e}x1000",
           lambda: PCA(n_components=50, svd_solver="full").fit(Xw))
    del Xw

    from sklearn.linear_model import SGDClassifier
    from mi355ml.model_selection import HyperbandSearchCV
    Xs, ys = datasets.make_classification(n_samples=1_000_000,
                                          n_features=50, n_informative=10,
                                          random_state=3)
    with mi355ml.config.set({"array.block_rows": 100_000}):
        hb = timeit("Hyperband(SGDClassifier CPU models, max_iter=27) 1e6x50",
                    lambda: HyperbandSearchCV(
                        SGDClassifier(loss="log_loss", random_state=0),
                        {"alpha": [10 ** -i for i in range(2, 7)],
                         "eta0": [0.01, 0.1, 1.0],
                         "learning_rate": ["constant", "invscaling"]},
                        max_iter=27, random_state=0).fit(Xs, ys))
    print(f"    best={hb.best_score_:.4f} models={hb.metadata_['n_models']} "
          f"pf_calls={hb.metadata_['partial_fit_calls']}")


if __name__ == "__main__":
    main()
