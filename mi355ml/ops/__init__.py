"""Per-shard compute ops.

Every op here is the MI355X replacement of one per-chunk compute function the
reference's graphs launch (SURVEY §2.7 kernel table K1–K20). Each op has:

  - a hand-written CDNA4 HIP kernel (mi355ml/_hip, built for gfx950) — the
    ONLY path for CUDA tensors; if the extension is missing on a GPU box the
    op raises rather than silently running eager torch
    (config "kernels.require_hip"),
  - a plain PyTorch CPU implementation, which doubles as the fp32 numerics
    oracle for the HIP kernel tests.

There is deliberately NO pluggable multi-backend dispatch (the reference's
dask/array/dispatch.py design is rejected per SURVEY §1 L0): cuda -> HIP,
cpu -> torch, nothing else.
"""

from .reductions import col_moments, col_minmax, col_sum, merge_moments
from .glm import glm_value_grad
from .kmeans import kmeans_assign, closest_center
from .elementwise import affine_transform

__all__ = [
    "col_moments",
    "col_minmax",
    "col_sum",
    "merge_moments",
    "glm_value_grad",
    "kmeans_assign",
    "closest_center",
    "affine_transform",
]
