"""Break down KMeans k-means|| init time on the config-3 shard shape.

Instruments each stage of k_init (first d2 pass, per-round refine/sample,
final candidate labeling, sklearn local KMeans) with device-synchronized
wall clocks to find where the ~2.4 s goes.
"""

import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))
import numpy as np
import torch

import mi355ml.cluster.k_means as km
from mi355ml.core.array import from_numpy
from mi355ml import datasets


def main(m=125_000_000 // 4, d=50, k=100):
    # quarter-size shard keeps the probe cheap; costs scale linearly in m
    X, _ = datasets.make_blobs(n_samples=m, n_features=d, centers=k,
                               random_state=0)
    torch.cuda.synchronize()

    T = {}

    def clock(name, fn, *a, **kw):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        out = fn(*a, **kw)
        torch.cuda.synchronize()
        T[name] = T.get(name, 0.0) + time.perf_counter() - t0
        return out

    orig_closest = km.closest_center
    calls = {"closest": 0}

    def timed_closest(blk, cset):
        calls["closest"] += 1
        return clock(f"closest(k={cset.shape[0]})", orig_closest, blk, cset)

    km.closest_center = timed_closest
    import sklearn.cluster as skc
    orig_fit = skc.KMeans.fit

    def timed_fit(self, *a, **kw):
        return clock("sklearn_local_kmeans", orig_fit, self, *a, **kw)

    skc.KMeans.fit = timed_fit

    orig_labels = km._shard_labels

    def timed_labels(X_, c):
        return clock(f"final_labels(cands={c.shape[0]})", orig_labels, X_, c)

    km._shard_labels = timed_labels

    torch.cuda.synchronize()
    t0 = time.perf_counter()
    centers = km.k_init(X, k, random_state=0)
    torch.cuda.synchronize()
    total = time.perf_counter() - t0
    print(f"k_init total: {total:.3f} s  ({m}x{d}, k={k}, "
          f"{calls['closest']} closest calls)")
    acc = 0.0
    for name, t in sorted(T.items(), key=lambda kv: -kv[1]):
        print(f"  {name:34s} {t:7.3f} s")
        acc += t
    print(f"  {'(untimed remainder)':34s} {total - acc:7.3f} s")
    assert centers.shape == (k, d)


if __name__ == "__main__":
    main()
