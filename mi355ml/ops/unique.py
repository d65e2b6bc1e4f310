"""Distributed unique / value counts (K14).

Per-shard torch.unique + one ragged all-gather + local merge — the
reference's set-union tree (dask/array/routines.py:1659, bag foldby)
collapsed to a single collective.
"""

from __future__ import annotations

from typing import Tuple

import torch

from ..core.array import DistArray


def unique(X: DistArray, return_counts: bool = False):
    """Sorted global unique values of a 1-D (or flattened) DistArray."""
    local = X.local.reshape(-1)
    if return_counts:
        lv, lc = torch.unique(local, return_counts=True)
        if X.comm.world_size > 1:
            vs = torch.cat(X.comm.all_gather_varlen(lv.contiguous()))
            cs = torch.cat(X.comm.all_gather_varlen(lc.contiguous()))
        else:
            vs, cs = lv, lc
        gv, inv = torch.unique(vs, return_inverse=True)
        gc = torch.zeros(gv.shape[0], dtype=cs.dtype, device=cs.device)
        gc.index_add_(0, inv, cs)
        return gv, gc
    lv = torch.unique(local)
    if X.comm.world_size > 1:
        vs = torch.cat(X.comm.all_gather_varlen(lv.contiguous()))
        return torch.unique(vs)
    return lv


def column_uniques(X: DistArray, max_categories: int = 1 << 20):
    """Per-column sorted unique values (for encoders). Returns a list of
    1-D tensors, replicated."""
    x = X._as2d()
    outs = []
    for c in range(x.shape[1]):
        col = DistArray(x[:, c].contiguous(), X.global_rows, X.row_offset,
                        X.comm)
        u = unique(col)
        if u.shape[0] > max_categories:
            raise ValueError(f"column {c} has {u.shape[0]} categories")
        outs.append(u)
    return outs


def bincount(X: DistArray, minlength: int = 0) -> torch.Tensor:
    """Global bincount of a non-negative integer DistArray (allreduce)."""
    local = X.local.reshape(-1).long()
    n = int(local.max().item()) + 1 if local.numel() else 0
    nt = torch.tensor([max(n, minlength)], dtype=torch.int64,
                      device=local.device)
    if X.comm.world_size > 1:
        X.comm.all_reduce(nt, op="max")
    length = int(nt.item())
    c = torch.bincount(local, minlength=length).to(torch.float64)
    if X.comm.world_size > 1:
        X.comm.all_reduce(c)
    return c.long()
