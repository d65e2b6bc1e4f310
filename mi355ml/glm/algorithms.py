"""Distributed GLM solvers (dask_glm/algorithms.py parity — SURVEY §2.2, §3.2).

SPMD design: every rank runs the SAME driver loop (scipy lbfgs, Armijo GD,
Newton, ADMM, proximal-grad); per-iteration distributed work is ONE fused
objective+gradient kernel pass over the local shard (ops.glm_value_grad)
plus ONE small fused-buffer all-reduce ([loss, grad_b, grad_w] — SURVEY C1/C2
"fuse stats into one buffer per iteration"). The driver-side direction
computation is redundant-deterministic across ranks, so no broadcasts are
needed (SURVEY §3.2 note).

Conventions: w is the [d] coefficient vector, b the (unpenalized) intercept;
lamduh is the regularization weight λ (dask_glm naming).
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import numpy as np
import torch
from scipy.optimize import fmin_l_bfgs_b

from ..core.array import DistArray
from ..ops import glm_value_grad
from .families import get_family
from .regularizers import get_regularizer


def _shard_value_grad(X: DistArray, y: DistArray, w: torch.Tensor, b: float,
                      family: str):
    """Per-shard fused (loss, gw, gb); streams blocks for spilled shards."""
    from ..core.sparse import SparseDistArray
    from ..core.spill import SpilledDistArray
    if isinstance(X, SparseDistArray):
        from ..ops.glm import sparse_value_grad
        return sparse_value_grad(X, y.local, w, b, family)
    if isinstance(X, SpilledDistArray):
        loss = gw = gb = None
        for lo, blk in X.device_blocks():
            yb = y.local[lo:lo + blk.shape[0]]
            if yb.device != blk.device:
                yb = yb.to(blk.device, non_blocking=True)
            l_, gw_, gb_ = glm_value_grad(blk, yb, w.to(blk.device),
                                          torch.tensor(b), family)
            if loss is None:
                loss, gw, gb = l_, gw_, gb_
            else:
                loss, gw, gb = loss + l_, gw + gw_, gb + gb_
        return loss, gw, gb
    return glm_value_grad(X.local, y.local, w, torch.tensor(b), family)


def _dist_value_grad(X: DistArray, y: DistArray, w: torch.Tensor, b: float,
                     family: str) -> Tuple[float, torch.Tensor, float]:
    """Global (loss, grad_w, grad_b) — fused kernel + one fused allreduce."""
    from ..core.colarray import ColDistArray
    if isinstance(X, ColDistArray):
        # feature-parallel path: col_value_grad does its own collectives
        from ..ops.glm import col_value_grad
        yt = y.local if hasattr(y, "local") else y
        loss, gw, gb = col_value_grad(X, yt, w, b, family)
        return float(loss), gw, float(gb)
    loss, gw, gb = _shard_value_grad(X, y, w, b, family)
    buf = torch.empty(2 + gw.shape[0], dtype=torch.float64, device=gw.device)
    buf[0] = loss
    buf[1] = gb
    buf[2:] = gw
    if X.comm.world_size > 1:
        X.comm.all_reduce(buf)
    return float(buf[0]), buf[2:], float(buf[1])


def _local_value_grad(X: DistArray, y: DistArray, w: torch.Tensor, b: float,
                      family: str) -> Tuple[float, torch.Tensor, float]:
    from ..core.colarray import ColDistArray
    if isinstance(X, ColDistArray):
        raise ValueError(
            "admm needs per-rank row shards (local subproblems); "
            "column-sharded X supports lbfgs/gradient_descent/"
            "proximal_grad — SURVEY §5.7")
    loss, gw, gb = _shard_value_grad(X, y, w, b, family)
    return float(loss), gw, float(gb)


def lbfgs(X: DistArray, y: DistArray, family="logistic", lamduh: float = 0.0,
          max_iter: int = 100, tol: float = 1e-4, fit_intercept: bool = True,
          verbose: bool = False) -> Tuple[torch.Tensor, float, int]:
    """L-BFGS with smooth (L2) penalty — the BASELINE LogReg solver
    (reference: dask_glm lbfgs via scipy.optimize.fmin_l_bfgs_b).
    """
    fam = get_family(family).name
    d = X.ncols
    dev = X.device
    n_evals = [0]

    def objective(theta: np.ndarray):
        w = torch.from_numpy(theta[:d]).to(device=dev, dtype=torch.float32)
        b = float(theta[d]) if fit_intercept else 0.0
        loss, gw, gb = _dist_value_grad(X, y, w, b, fam)
        n_evals[0] += 1
        if lamduh > 0.0:
            wd = w.double()
            loss = loss + 0.5 * lamduh * float((wd * wd).sum())
            gw = gw + lamduh * wd
        g = np.empty(d + 1)
        g[:d] = gw.cpu().numpy()
        g[d] = gb if fit_intercept else 0.0
        return loss, g

    x0 = np.zeros(d + 1)
    xopt, fval, info = fmin_l_bfgs_b(
        objective, x0, fprime=None, maxiter=max_iter,
        pgtol=tol, factr=1e7)
    w = torch.from_numpy(xopt[:d]).to(device=dev, dtype=torch.float32)
    b = float(xopt[d]) if fit_intercept else 0.0
    return w, b, int(info.get("nit", n_evals[0]))


def gradient_descent(X: DistArray, y: DistArray, family="logistic",
                     max_iter: int = 100, tol: float = 1e-8,
                     fit_intercept: bool = True, **kw
                     ) -> Tuple[torch.Tensor, float, int]:
    """Armijo backtracking gradient descent (dask_glm gradient_descent)."""
    fam = get_family(family).name
    d = X.ncols
    dev = X.device
    w = torch.zeros(d, dtype=torch.float32, device=dev)
    b = 0.0
    stepSize = 1.0
    armijoMult = 0.1
    backtrackMult = 0.5
    stepGrowth = 1.25
    loss, gw, gb = _dist_value_grad(X, y, w, b, fam)
    it = 0
    for it in range(max_iter):
        gnorm2 = float((gw * gw).sum()) + (gb * gb if fit_intercept else 0.0)
        # backtracking line search on the fused objective
        for _ in range(100):
            w_new = (w.double() - stepSize * gw).float()
            b_new = b - stepSize * gb if fit_intercept else 0.0
            new_loss, new_gw, new_gb = _dist_value_grad(X, y, w_new, b_new, fam)
            if new_loss <= loss - armijoMult * stepSize * gnorm2:
                break
            stepSize *= backtrackMult
        else:
            break
        converged = abs(loss - new_loss) < tol * max(abs(loss), 1.0)
        w, b, loss, gw, gb = w_new, b_new, new_loss, new_gw, new_gb
        if converged:
            break
        stepSize *= stepGrowth
    return w, b, it + 1


def newton(X: DistArray, y: DistArray, family="logistic", max_iter: int = 50,
           tol: float = 1e-8, fit_intercept: bool = True, **kw
           ) -> Tuple[torch.Tensor, float, int]:
    """Newton's method: H = XᵀDX via blocked per-shard syrk + allreduce
    (dask_glm newton; SURVEY §2.2 — driver-side lstsq on the small H)."""
    from ..core.colarray import ColDistArray
    if isinstance(X, ColDistArray):
        raise ValueError("newton builds a dense [d+1,d+1] Hessian — not "
                         "supported for column-sharded (wide) X; use lbfgs")
    fam_cls = get_family(family)
    fam = fam_cls.name
    d = X.ncols
    dev = X.device
    da = d + (1 if fit_intercept else 0)
    w = torch.zeros(d, dtype=torch.float32, device=dev)
    b = 0.0
    it = 0
    for it in range(max_iter):
        loss, gw, gb = _dist_value_grad(X, y, w, b, fam)
        H = torch.zeros(da, da, dtype=torch.float64, device=dev)
        for blk in X.blocks():
            z = blk @ w + b
            h = fam_cls.hessian_weights(z.double())
            xb = blk.double()
            xh = xb * h[:, None]
            H[:d, :d] += xb.t() @ xh
            if fit_intercept:
                s = xh.sum(dim=0)
                H[:d, d] += s
                H[d, :d] += s
                H[d, d] += h.sum()
        if X.comm.world_size > 1:
            X.comm.all_reduce(H)
        g = torch.empty(da, dtype=torch.float64, device=dev)
        g[:d] = gw
        if fit_intercept:
            g[d] = gb
        step = torch.linalg.lstsq(H.cpu(), g.cpu()[:, None]).solution[:, 0]
        step = step.to(dev)
        w = (w.double() - step[:d]).float()
        if fit_intercept:
            b = b - float(step[d])
        if float(step.abs().max()) < tol:
            break
    return w, b, it + 1


def proximal_grad(X: DistArray, y: DistArray, family="logistic",
                  regularizer="l1", lamduh: float = 0.1, max_iter: int = 100,
                  tol: float = 1e-7, fit_intercept: bool = True, **kw
                  ) -> Tuple[torch.Tensor, float, int]:
    """Proximal gradient with backtracking (dask_glm proximal_grad)."""
    fam = get_family(family).name
    reg = get_regularizer(regularizer)
    d = X.ncols
    dev = X.device
    w = torch.zeros(d, dtype=torch.float32, device=dev)
    b = 0.0
    stepSize = 1.0
    backtrackMult = 0.5
    stepGrowth = 1.25
    loss, gw, gb = _dist_value_grad(X, y, w, b, fam)
    it = 0
    for it in range(max_iter):
        ok = False
        for _ in range(100):
            w_new = reg.proximal_operator(
                (w.double() - stepSize * gw), lamduh * stepSize).float()
            b_new = b - stepSize * gb if fit_intercept else 0.0
            dw = (w_new.double() - w.double())
            db = b_new - b
            new_loss, new_gw, new_gb = _dist_value_grad(X, y, w_new, b_new, fam)
            quad = (loss + float((gw * dw).sum()) + gb * db
                    + (float((dw * dw).sum()) + db * db) / (2.0 * stepSize))
            if new_loss <= quad:
                ok = True
                break
            stepSize *= backtrackMult
        if not ok:
            break
        shift = float((w_new.double() - w.double()).abs().max())
        w, b, loss, gw, gb = w_new, b_new, new_loss, new_gw, new_gb
        if shift < tol:
            break
        stepSize *= stepGrowth
    return w, b, it + 1


def admm(X: DistArray, y: DistArray, family="logistic", regularizer="l1",
         lamduh: float = 0.1, rho: float = 1.0, max_iter: int = 100,
         abstol: float = 1e-4, reltol: float = 1e-2,
         fit_intercept: bool = True, **kw) -> Tuple[torch.Tensor, float, int]:
    """Consensus ADMM: one local subproblem per rank-shard, z-update = prox of
    the averaged local β (the only communication: one small allreduce per
    round — dask_glm admm / SURVEY §2.2)."""
    fam = get_family(family).name
    reg = get_regularizer(regularizer)
    d = X.ncols
    dev = X.device
    da = d + (1 if fit_intercept else 0)
    nranks = X.comm.world_size
    beta = np.zeros(da)
    z = torch.zeros(da, dtype=torch.float64, device=dev)
    u = torch.zeros(da, dtype=torch.float64, device=dev)

    def local_obj(theta: np.ndarray, z_np: np.ndarray, u_np: np.ndarray):
        w = torch.from_numpy(theta[:d]).to(device=dev, dtype=torch.float32)
        b = float(theta[d]) if fit_intercept else 0.0
        loss, gw, gb = _local_value_grad(X, y, w, b, fam)
        diff = theta - z_np + u_np
        loss = loss + 0.5 * rho * float(diff @ diff)
        g = np.empty(da)
        g[:d] = gw.cpu().numpy()
        if fit_intercept:
            g[d] = gb
        g += rho * diff
        return loss, g

    it = 0
    for it in range(max_iter):
        z_np = z.cpu().numpy()
        u_np = u.cpu().numpy()
        beta, _, _ = fmin_l_bfgs_b(local_obj, beta, args=(z_np, u_np),
                                   maxiter=50, pgtol=1e-7)
        bt = torch.from_numpy(beta).to(device=dev, dtype=torch.float64)
        # consensus average of (beta + u) across ranks
        avg = (bt + u).clone()
        if nranks > 1:
            X.comm.all_reduce(avg)
            avg /= nranks
        z_old = z.clone()
        # prox applies to coefficients only; intercept is averaged untouched
        zw = reg.proximal_operator(avg[:d].clone(), lamduh / (rho * nranks))
        z = avg.clone()
        z[:d] = zw
        u = u + bt - z
        # stopping: primal/dual residuals (boyd)
        r2 = float(((bt - z) ** 2).sum())
        if nranks > 1:
            t = torch.tensor([r2], dtype=torch.float64, device=dev)
            X.comm.all_reduce(t)
            r2 = float(t[0])
        prires = math.sqrt(r2)
        duares = rho * math.sqrt(nranks) * float((z - z_old).norm())
        eps_pri = math.sqrt(da * nranks) * abstol + reltol * max(
            float(bt.norm()) * math.sqrt(nranks), float(z.norm()))
        eps_dua = math.sqrt(da * nranks) * abstol + reltol * rho * float(u.norm())
        if prires < eps_pri and duares < eps_dua:
            break
    w = z[:d].float()
    b = float(z[d]) if fit_intercept else 0.0
    return w, b, it + 1


SOLVERS = {
    "lbfgs": lbfgs,
    "gradient_descent": gradient_descent,
    "newton": newton,
    "proximal_grad": proximal_grad,
    "admm": admm,
}
