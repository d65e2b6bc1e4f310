"""Fused per-column affine transform (K9 in SURVEY §2.7).

out[i, j] = x[i, j] * scale[j] + shift[j], one pass, vectorized.
Covers StandardScaler ((x−μ)/σ: scale=1/σ, shift=−μ/σ), MinMaxScaler and
RobustScaler transforms and their inverses. The reference builds these as
elemwise blockwise graphs (dask/array/core.py:4387) with one temporary per
arithmetic op; the HIP kernel is a single read-modify-write at HBM bandwidth.
"""

from __future__ import annotations

import torch

from ._ext import hip_ext, use_hip


def affine_transform(x: torch.Tensor, scale: torch.Tensor,
                     shift: torch.Tensor, out: torch.Tensor = None
                     ) -> torch.Tensor:
    """x * scale[col] + shift[col]; optionally into `out` (may alias x)."""
    assert x.dim() == 2
    assert scale.shape == (x.shape[1],) and shift.shape == (x.shape[1],)
    if use_hip(x):
        ext = hip_ext()
        if out is None:
            out = torch.empty_like(x)
        ext.affine_transform(x.contiguous(), scale.contiguous(),
                             shift.contiguous(), out)
        return out
    res = x * scale[None, :] + shift[None, :]
    if out is not None:
        out.copy_(res)
        return out
    return res
