"""Fused GLM objective+gradient op (K1+K2+K3 in SURVEY §2.7).

The reference evaluates a GLM objective as three separate dask graphs per
lbfgs iteration — Xβ tensordot, pointwise family ops, Xᵀr tensordot + sum
tree — i.e. ≥2 full passes over X plus graph overhead. Here the
whole objective callback is ONE fused HIP kernel pass over the shard
(X staged once through LDS, used for both z=X·w and the g=Xᵀr partial), so
the per-iteration cost is bytes(X)/HBM-bandwidth.

Families (dask_glm/families.py parity): logistic, normal, poisson.
  logistic: loss=softplus(z)-y·z            r=sigmoid(z)-y
  normal:   loss=0.5·(z-y)²                 r=z-y
  poisson:  loss=exp(z)-y·z                 r=exp(z)-y
(z = x·w + b; loss summed over rows; grad_w = Σ r·x, grad_b = Σ r.)
"""

from __future__ import annotations

from typing import Tuple

import torch

from ._ext import hip_ext, use_hip

FAMILIES = ("logistic", "normal", "poisson")
_FAMILY_ID = {f: i for i, f in enumerate(FAMILIES)}


def glm_value_grad(x: torch.Tensor, y: torch.Tensor, w: torch.Tensor,
                   b: torch.Tensor, family: str = "logistic",
                   ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Per-shard (loss_sum, grad_w, grad_b), all float64.

    x: [m, d] f32; y: [m] f32; w: [d] f32; b: scalar f32 tensor.
    """
    assert family in _FAMILY_ID, family
    assert x.dim() == 2 and y.dim() == 1 and w.dim() == 1
    if use_hip(x):
        if x.shape[1] <= 512:
            ext = hip_ext()
            loss, gw, gb = ext.glm_value_grad(
                x.contiguous(), y.contiguous(), w.contiguous(),
                float(b), _FAMILY_ID[family])
            return loss, gw, gb
        # wide-feature path (d > the fused kernel's LDS tile): explicit
        # rocBLAS-composed GEMV pair (plain library GEMMs, sanctioned) with
        # the link math in f32 on device and f64 partials.
        return _composed_value_grad(x, y, w, float(b), family)
    xd = x.to(torch.float64)
    yd = y.to(torch.float64)
    wd = w.to(torch.float64)
    z = xd @ wd + float(b)
    if family == "logistic":
        loss = (torch.nn.functional.softplus(z) - yd * z).sum()
        r = torch.sigmoid(z) - yd
    elif family == "normal":
        diff = z - yd
        loss = 0.5 * (diff * diff).sum()
        r = diff
    else:  # poisson
        ez = torch.exp(z)
        loss = (ez - yd * z).sum()
        r = ez - yd
    gw = xd.t() @ r
    gb = r.sum()
    return loss, gw, gb


def sparse_value_grad(X, y_local: torch.Tensor, w: torch.Tensor, b: float,
                      family: str):
    """Per-shard (loss, gw, gb) for a SparseDistArray shard (CSR SpMV pair,
    f64 link math — the encoders' sparse one-hot output feeds the GLM
    solvers without densifying)."""
    z = X.dot(w.to(torch.float64)).local.double() + float(b)
    yd = y_local.to(device=z.device, dtype=torch.float64)
    if family == "logistic":
        loss = (torch.nn.functional.softplus(z) - yd * z).sum()
        r = torch.sigmoid(z) - yd
    elif family == "normal":
        diff = z - yd
        loss = 0.5 * (diff * diff).sum()
        r = diff
    else:  # poisson
        ez = torch.exp(z)
        loss = (ez - yd * z).sum()
        r = ez - yd
    gw = X._xt_dot_local(r, torch.float64)
    gb = r.sum()
    return loss, gw, gb


def _composed_value_grad(x, y, w, b, family):
    z = x @ w + b
    if family == "logistic":
        loss = (torch.nn.functional.softplus(z).double()
                - y.double() * z.double()).sum()
        r = torch.sigmoid(z) - y
    elif family == "normal":
        diff = (z - y)
        loss = 0.5 * (diff.double() * diff.double()).sum()
        r = diff
    else:  # poisson
        ez = torch.exp(z.double())
        loss = (ez - y.double() * z.double()).sum()
        r = (ez - y.double()).float()
    gw = (x.t() @ r).double()
    gb = r.double().sum()
    return loss, gw, gb


def glm_predict_linear(x: torch.Tensor, w: torch.Tensor, b: float
                       ) -> torch.Tensor:
    """z = X·w + b for a shard (library GEMV; not a hot op)."""
    return x @ w.to(x.dtype) + b


def col_value_grad(X, y_full: torch.Tensor, w_full: torch.Tensor, b: float,
                   family: str):
    """Feature-parallel fused objective for a column-sharded X
    (core.colarray.ColDistArray — SURVEY §5.7 wide-X mechanism).

    z-pass: partial GEMV over the local column slice + ONE [m] allreduce
    (the feature-parallel dual of the row-shard's local z). Pointwise link
    math runs replicated (identical on every rank); the gradient slice
    Xᵀr is LOCAL (columns already sharded) and is allgathered into the
    full [d] vector the redundant-deterministic lbfgs driver needs.

    Returns GLOBAL (loss, grad_w[d] f64, grad_b) — callers must NOT
    allreduce again.
    """
    assert family in _FAMILY_ID, family
    yd = y_full.to(device=X.device, dtype=torch.float64)
    z = X.dot(w_full.to(X.dtype))          # replicated [m] (one allreduce)
    zd = z.double() + float(b)
    if family == "logistic":
        loss = (torch.nn.functional.softplus(zd) - yd * zd).sum()
        r = torch.sigmoid(zd) - yd
    elif family == "normal":
        diff = zd - yd
        loss = 0.5 * (diff * diff).sum()
        r = diff
    else:  # poisson
        ez = torch.exp(zd)
        loss = (ez - yd * zd).sum()
        r = ez - yd
    gl = X.rmatvec_local(r.to(X.dtype)).double()   # [d_local], no comm
    gw = X.gather_cols(gl)                          # full [d]
    gb = r.sum()
    return loss, gw, gb
