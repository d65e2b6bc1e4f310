"""Column-statistics ops (K7/K8 in SURVEY §2.7).

The reference computes these as chunk/combine/aggregate triples tree-reduced
through the task graph (dask/array/reductions.py:581-824 moment math,
:336-543 min/max/sum). Here each is ONE fused pass over the shard (HIP kernel
on GPU) producing a small per-column partial; the cross-rank combine is the
exact Chan/pairwise moment-merge applied to allgathered partials.

All accumulators are float64 (fp32 data, fp64 accumulation — SURVEY A.2).
"""

from __future__ import annotations

from typing import Tuple

import torch

from ._ext import hip_ext, use_hip


def merge_moments(ns: torch.Tensor, sums: torch.Tensor, m2s: torch.Tensor
                  ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Merge G groups of {n, sum, M2} into one (exact one-level Chan merge).

    ns: [G] float64; sums, m2s: [G, F] float64.
    M2_total = sum_i M2_i + sum_i n_i * (mean_i - mean)^2
    (reference math: dask/array/reductions.py:690-785).
    """
    n = ns.sum()
    total = sums.sum(dim=0)
    if float(n) == 0.0:
        return n, total, torch.zeros_like(total)
    gmean = total / n
    nz = ns > 0
    means = torch.where(nz[:, None], sums / ns.clamp(min=1.0)[:, None],
                        torch.zeros_like(sums))
    delta = means - gmean
    m2 = m2s.sum(dim=0) + (ns[:, None] * delta * delta * nz[:, None]).sum(dim=0)
    return n, total, m2


def col_moments(x: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """One-pass per-column {n, sum, M2} of a 2-D shard.

    Returns (n [scalar f64], sum [F f64], M2 [F f64]).
    """
    assert x.dim() == 2
    m = x.shape[0]
    if use_hip(x):
        ext = hip_ext()
        ns, sums, m2s = ext.col_moments_partial(x.contiguous())
        return merge_moments(ns, sums, m2s)
    xd = x.to(torch.float64)
    n = torch.tensor(float(m), dtype=torch.float64, device=x.device)
    s = xd.sum(dim=0)
    if m == 0:
        return n, s, torch.zeros_like(s)
    mean = s / m
    m2 = ((xd - mean) ** 2).sum(dim=0)
    return n, s, m2


def col_minmax(x: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """Per-column (min, max) of a 2-D shard; returned in x.dtype."""
    assert x.dim() == 2
    if x.shape[0] == 0:
        f = x.shape[1]
        return (torch.full((f,), float("inf"), dtype=x.dtype, device=x.device),
                torch.full((f,), float("-inf"), dtype=x.dtype, device=x.device))
    if use_hip(x):
        ext = hip_ext()
        mins, maxs = ext.col_minmax_partial(x.contiguous())
        return mins.amin(dim=0), maxs.amax(dim=0)
    return x.amin(dim=0), x.amax(dim=0)


def col_sum(x: torch.Tensor) -> torch.Tensor:
    """Per-column float64 sum of a 2-D shard."""
    assert x.dim() == 2
    if use_hip(x):
        ext = hip_ext()
        ns, sums, _ = ext.col_moments_partial(x.contiguous())
        return sums.sum(dim=0)
    return x.sum(dim=0, dtype=torch.float64)
