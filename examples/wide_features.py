"""Wide-X (feature-sharded) training: 1e5 rows x 50k columns.

Row-sharding is wrong for very wide X (the per-iteration w broadcast and
X^T r reduction would dominate); ColDistArray shards the FEATURE axis:
every rank holds all rows and a column slice.

    python -m mi355ml.parallel.launch --gpus 8 examples/wide_features.py
"""
import os
import sys

import torch

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))
import mi355ml  # noqa: F401,E402
from mi355ml.core.colarray import ColDistArray  # noqa: E402
from mi355ml.decomposition import PCA, TruncatedSVD  # noqa: E402
from mi355ml.linear_model import LogisticRegression  # noqa: E402
from mi355ml.parallel import comm as comm_mod  # noqa: E402


def main():
    comm = comm_mod.init()
    on_gpu = comm.device.type == "cuda"
    n, d = (100_000, 50_000) if on_gpu else (2_000, 1_000)
    lo, hi = (d * comm.rank) // comm.world_size, \
        (d * (comm.rank + 1)) // comm.world_size
    g = torch.Generator(device=comm.device).manual_seed(7 + comm.rank)
    # each rank generates ITS column slice (no full X anywhere)
    x_local = torch.randn(n, hi - lo, generator=g, device=comm.device)
    X = ColDistArray(x_local, d, lo, comm)
    # labels from a sparse true weight vector on the first columns
    w_true = torch.zeros(d, device=comm.device)
    if lo == 0:
        w_true[:64] = torch.randn(64, generator=g, device=comm.device)
    z = X.dot(w_true)  # partial GEMV + one [n] allreduce
    y = (torch.sigmoid(z) > 0.5).float()

    model = LogisticRegression(solver="lbfgs", max_iter=30).fit(X, y)
    acc = model.score(X, y.cpu().numpy())
    if comm.rank == 0:
        print(f"wide-X LogReg acc={acc:.3f}")

    tsvd = TruncatedSVD(n_components=20, algorithm="randomized").fit(X)
    pca = PCA(n_components=20, svd_solver="randomized").fit(X)
    if comm.rank == 0:
        print("tsvd sv[:3]:", tsvd.singular_values_[:3].tolist())
        print("pca  sv[:3]:", pca.singular_values_[:3].tolist())
    comm_mod.shutdown()


if __name__ == "__main__":
    main()
