"""Small utilities (dask_ml/utils.py & _utils.py parity)."""

from __future__ import annotations

import torch

from .base import check_array, check_X_y, check_is_fitted  # noqa: F401


def svd_flip(u: torch.Tensor, v: torch.Tensor, u_based_decision: bool = False):
    """Deterministic SVD sign convention (sklearn/utils/extmath.svd_flip).
    The V-based form is used distributed (V is small and replicated)."""
    if u_based_decision:
        idx = u.abs().argmax(dim=0)
        signs = torch.sign(u[idx, torch.arange(u.shape[1], device=u.device)])
    else:
        idx = v.abs().argmax(dim=1)
        signs = torch.sign(v[torch.arange(v.shape[0], device=v.device), idx])
    signs[signs == 0] = 1.0
    return u * signs[None, :], v * signs[:, None]


def handle_zeros_in_scale(scale: torch.Tensor) -> torch.Tensor:
    out = scale.clone()
    out[out == 0.0] = 1.0
    return out


def check_chunks(n_samples: int, n_features: int, chunks=None) -> tuple:
    """Pick a row-block size (reference check_chunks analog): aim for ~2x
    the rank count of blocks, >= 100 rows each."""
    if chunks is not None:
        return chunks
    from .parallel import comm as comm_mod
    w = comm_mod.default().world_size
    rows = max(100, n_samples // max(2 * w, 1))
    return (rows, n_features)


def draw_seed(rng, low: int = 0, high: int = 2 ** 31 - 1) -> int:
    """Draw a child seed from an int / RandomState / Generator."""
    import numpy as np
    rs = np.random.RandomState(rng) if isinstance(rng, (int, type(None))) \
        else rng
    if hasattr(rs, "randint"):        # RandomState
        return int(rs.randint(low, high))
    return int(rs.integers(low, high))  # np.random.Generator


def slice_columns(X, columns):
    """Column subset of a DistArray (local slice, no communication)."""
    return X.map_local(lambda t: t[:, columns])
