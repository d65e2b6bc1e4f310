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


def pairwise_distances(X, Y: torch.Tensor, metric: str = "euclidean"):
    """Row-sharded distances from X to a small replicated Y
    (dask_ml.metrics.pairwise_distances parity for dense metrics)."""
    Xd = check_array(X)
    Y = torch.as_tensor(Y, dtype=Xd.dtype, device=Xd.device)

    def _block(t: torch.Tensor) -> torch.Tensor:
        if metric in ("euclidean", "sqeuclidean"):
            xn = (t.double() * t.double()).sum(dim=1)
            yn = (Y.double() * Y.double()).sum(dim=1)
            d2 = (xn[:, None] + yn[None, :]
                  - 2.0 * (t.double() @ Y.double().t())).clamp(min=0.0)
            out = d2 if metric == "sqeuclidean" else d2.sqrt()
        elif metric == "cosine":
            a = t.double() / t.double().norm(dim=1, keepdim=True).clamp(1e-12)
            b = Y.double() / Y.double().norm(dim=1, keepdim=True).clamp(1e-12)
            out = 1.0 - a @ b.t()
        elif metric in ("manhattan", "cityblock", "l1"):
            out = (t.double()[:, None, :] - Y.double()[None, :, :]) \
                .abs().sum(dim=2)
        else:
            raise ValueError(f"unsupported metric {metric!r}")
        return out.to(t.dtype)

    return Xd.map_local(_block)


def linear_kernel(X, Y: torch.Tensor):
    Xd = check_array(X)
    Y = torch.as_tensor(Y, dtype=Xd.dtype, device=Xd.device)
    return Xd.map_local(lambda t: t @ Y.t())


def rbf_kernel(X, Y: torch.Tensor, gamma=None):
    Xd = check_array(X)
    Y = torch.as_tensor(Y, dtype=Xd.dtype, device=Xd.device)
    g = gamma if gamma is not None else 1.0 / Xd.ncols

    def _k(t):
        xn = (t * t).sum(dim=1)
        yn = (Y * Y).sum(dim=1)
        d2 = (xn[:, None] + yn[None, :] - 2.0 * (t @ Y.t())).clamp(min=0)
        return torch.exp(-g * d2)

    return Xd.map_local(_k)


def polynomial_kernel(X, Y: torch.Tensor, degree=3, gamma=None, coef0=1.0):
    Xd = check_array(X)
    Y = torch.as_tensor(Y, dtype=Xd.dtype, device=Xd.device)
    g = gamma if gamma is not None else 1.0 / Xd.ncols
    return Xd.map_local(lambda t: (g * (t @ Y.t()) + coef0) ** degree)


def sigmoid_kernel(X, Y: torch.Tensor, gamma=None, coef0=1.0):
    Xd = check_array(X)
    Y = torch.as_tensor(Y, dtype=Xd.dtype, device=Xd.device)
    g = gamma if gamma is not None else 1.0 / Xd.ncols
    return Xd.map_local(lambda t: torch.tanh(g * (t @ Y.t()) + coef0))
