"""Approximate distributed per-column percentiles (K13).

Per shard: an exact per-column quantile curve at the requested q plus the
{0,100} endpoints (torch.quantile). Merge (`merge_percentiles`,
dask/array/percentile.py:167 math): each shard's curve becomes
(value, count) pairs with count[j] = Δq[j]·N_shard, all pairs are sorted by
value, counts cumulated into a merged empirical CDF and the requested
quantiles interpolated. Same approximation class as the reference, so
RobustScaler / median imputation inherit identical semantics.
"""

from __future__ import annotations

from typing import Sequence

import torch

from ..core.array import DistArray


def _shard_curve(x: torch.Tensor, q_ext: torch.Tensor) -> torch.Tensor:
    """[nq, F] per-column quantiles of a 2-D shard (NaN-safe)."""
    if x.shape[0] == 0:
        return torch.full((q_ext.shape[0], x.shape[1]), float("nan"),
                          dtype=torch.float64, device=x.device)
    return torch.nanquantile(x.double(), q_ext.to(x.device) / 100.0, dim=0)


def merge_percentiles(qs: Sequence[float], curves: torch.Tensor,
                      ns: torch.Tensor, q_ext: torch.Tensor) -> torch.Tensor:
    """Merge G shard curves into global percentiles.

    curves: [G, nq_ext, F] float64; ns: [G] float64 (rows per shard);
    returns [len(qs), F] float64.
    """
    G, nq, F = curves.shape
    dev = curves.device
    dq = torch.empty(nq, dtype=torch.float64, device=dev)
    dq[0] = q_ext[0]
    dq[1:] = q_ext[1:] - q_ext[:-1]
    # (value, count) pairs: counts scaled per shard
    counts = (dq[None, :, None] / 100.0) * ns[:, None, None]  # [G, nq, 1]
    counts = counts.expand(G, nq, F).reshape(G * nq, F)
    vals = curves.reshape(G * nq, F)
    order = vals.argsort(dim=0)
    vals_s = torch.gather(vals, 0, order)
    counts_s = torch.gather(counts, 0, order)
    cum = counts_s.cumsum(dim=0)
    total = ns.sum()
    targets = torch.tensor(qs, dtype=torch.float64, device=dev) / 100.0 * total
    # batched per-column interpolation (one searchsorted over [F, G*nq])
    out = _interp_batched(targets.expand(F, -1), cum.t().contiguous(),
                          vals_s.t().contiguous())
    return out.t().contiguous()


def _interp(xq: torch.Tensor, xp: torch.Tensor, fp: torch.Tensor
            ) -> torch.Tensor:
    """np.interp equivalent (monotone xp)."""
    idx = torch.searchsorted(xp.contiguous(), xq, right=False)
    idx = idx.clamp(1, xp.shape[0] - 1)
    x0 = xp[idx - 1]
    x1 = xp[idx]
    y0 = fp[idx - 1]
    y1 = fp[idx]
    t = torch.where(x1 > x0, (xq - x0) / (x1 - x0),
                    torch.zeros_like(xq))
    res = y0 + t.clamp(0, 1) * (y1 - y0)
    res = torch.where(xq <= xp[0], fp[0], res)
    res = torch.where(xq >= xp[-1], fp[-1], res)
    return res


def _interp_batched(xq: torch.Tensor, xp: torch.Tensor, fp: torch.Tensor
                    ) -> torch.Tensor:
    """Row-batched np.interp: xq [F, M], xp/fp [F, N] (each row monotone).
    One batched searchsorted instead of a per-column Python loop."""
    idx = torch.searchsorted(xp, xq.contiguous(), right=False)
    idx = idx.clamp(1, xp.shape[1] - 1)
    x0 = torch.gather(xp, 1, idx - 1)
    x1 = torch.gather(xp, 1, idx)
    y0 = torch.gather(fp, 1, idx - 1)
    y1 = torch.gather(fp, 1, idx)
    t = torch.where(x1 > x0, (xq - x0) / (x1 - x0), torch.zeros_like(xq))
    res = y0 + t.clamp(0, 1) * (y1 - y0)
    res = torch.where(xq <= xp[:, :1], fp[:, :1].expand_as(res), res)
    res = torch.where(xq >= xp[:, -1:], fp[:, -1:].expand_as(res), res)
    return res


def col_percentiles(X: DistArray, qs: Sequence[float]) -> torch.Tensor:
    """Global per-column percentiles of a row-sharded array.

    Returns [len(qs), F] float64, replicated. Spilled shards contribute one
    per-BLOCK curve each (the merge handles any number of curves — the same
    approximation class as the reference's per-chunk summaries).
    """
    from ..core.spill import SpilledDistArray
    dev = X.device
    q_ext = sorted(set([0.0, 100.0] + [float(q) for q in qs]))
    q_ext_t = torch.tensor(q_ext, dtype=torch.float64)
    local_curves = []
    local_ns = []
    if isinstance(X, SpilledDistArray):
        for _, blk in X.device_blocks():
            b2 = blk if blk.dim() == 2 else blk[:, None]
            local_curves.append(_shard_curve(b2, q_ext_t))
            local_ns.append(float(b2.shape[0]))
    else:
        x = X._as2d()
        local_curves.append(_shard_curve(x, q_ext_t))
        local_ns.append(float(x.shape[0]))
    # pad to a fixed per-rank curve count for the dense all-gather
    nloc = torch.tensor([len(local_curves)], dtype=torch.int64, device=dev)
    if X.comm.world_size > 1:
        counts = [int(c.item()) for c in X.comm.all_gather(nloc)]
        gmax = max(counts)
        F = local_curves[0].shape[1]
        nq = len(q_ext)
        while len(local_curves) < gmax:
            local_curves.append(torch.full((nq, F), float("nan"),
                                           dtype=torch.float64, device=dev))
            local_ns.append(0.0)
        curves = torch.cat(X.comm.all_gather(
            torch.stack(local_curves).contiguous()))
        ns = torch.cat(X.comm.all_gather(
            torch.tensor(local_ns, dtype=torch.float64, device=dev)))
    else:
        curves = torch.stack(local_curves)
        ns = torch.tensor(local_ns, dtype=torch.float64, device=dev)
    keep = ns > 0
    curves, ns = curves[keep], ns[keep]
    merged = merge_percentiles([float(q) for q in qs], curves, ns,
                               q_ext_t.to(curves.device))
    return merged
