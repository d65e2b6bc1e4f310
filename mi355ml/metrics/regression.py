"""Regression metrics (dask_ml/metrics/regression.py parity).

Each metric is a per-shard fused reduction + one tiny allreduce (C1) —
the reference's blockwise + tree-reduce graphs collapsed.
"""

from __future__ import annotations

import torch

from ..base import check_array
from ..core.array import DistArray


def _pair(y_true, y_pred):
    yt = check_array(y_true, dtype=None)
    yp = check_array(y_pred, dtype=None)
    return yt, yp


def _reduce_sums(comm, *vals):
    t = torch.stack([v.double() for v in vals])
    if comm.world_size > 1:
        comm.all_reduce(t)
    return [float(v) for v in t]


def mean_squared_error(y_true, y_pred, squared: bool = True) -> float:
    yt, yp = _pair(y_true, y_pred)
    d = yt.local.double() - yp.local.double()
    s, n = _reduce_sums(yt.comm, (d * d).sum(),
                        torch.tensor(float(d.numel()), device=d.device))
    mse = s / max(n, 1.0)
    return mse if squared else mse ** 0.5


def mean_absolute_error(y_true, y_pred) -> float:
    yt, yp = _pair(y_true, y_pred)
    d = (yt.local.double() - yp.local.double()).abs()
    s, n = _reduce_sums(yt.comm, d.sum(),
                        torch.tensor(float(d.numel()), device=d.device))
    return s / max(n, 1.0)


def r2_score(y_true, y_pred) -> float:
    yt, yp = _pair(y_true, y_pred)
    t = yt.local.double()
    p = yp.local.double()
    dev = t.device
    ss_res = ((t - p) ** 2).sum()
    s = t.sum()
    s2 = (t * t).sum()
    n = torch.tensor(float(t.numel()), device=dev)
    ss_res, s, s2, n = _reduce_sums(yt.comm, ss_res, s, s2, n)
    mean = s / max(n, 1.0)
    ss_tot = s2 - n * mean * mean
    if ss_tot <= 1e-12 * max(s2, 1.0):
        # constant y_true: sklearn returns 1.0 for a perfect prediction,
        # 0.0 otherwise (advisor r1 low)
        return 1.0 if ss_res <= 1e-12 * max(s2, float(n)) else 0.0
    return 1.0 - ss_res / ss_tot
