"""Pairwise distances (dask_ml/metrics/pairwise.py parity).

euclidean_distances uses the ‖x‖²+‖y‖²−2xyᵀ expansion blockwise (the
reference's formulation); pairwise_distances_argmin_min is the fused
closest-center op (K4) — see ops/kmeans.py.
"""

from __future__ import annotations

import torch

from ..base import check_array
from ..core.array import DistArray
from ..ops import closest_center


def euclidean_distances(X, Y: torch.Tensor, squared: bool = False) -> DistArray:
    """Row-sharded distances from X (DistArray) to a small replicated Y."""
    Xd = check_array(X)
    Y = torch.as_tensor(Y, dtype=Xd.dtype, device=Xd.device)
    yn = (Y.double() * Y.double()).sum(dim=1)

    def _block(t: torch.Tensor) -> torch.Tensor:
        xn = (t.double() * t.double()).sum(dim=1)
        d2 = xn[:, None] + yn[None, :] - 2.0 * (t.double() @ Y.double().t())
        d2 = d2.clamp(min=0.0)
        return (d2 if squared else d2.sqrt()).to(t.dtype)

    return Xd.map_local(_block)


def pairwise_distances_argmin_min(X, Y: torch.Tensor):
    """(argmin labels, min Euclidean distance) of each X row against Y."""
    Xd = check_array(X)
    Y = torch.as_tensor(Y, dtype=Xd.dtype, device=Xd.device)
    labels, d2 = closest_center(Xd.local, Y)
    return (DistArray(labels, Xd.global_rows, Xd.row_offset, Xd.comm),
            DistArray(d2.sqrt(), Xd.global_rows, Xd.row_offset, Xd.comm))
