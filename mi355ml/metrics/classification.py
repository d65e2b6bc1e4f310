"""Classification metrics (dask_ml/metrics/classification.py parity)."""

from __future__ import annotations

import torch

from ..base import check_array


def accuracy_score(y_true, y_pred) -> float:
    yt = check_array(y_true, dtype=None)
    yp = check_array(y_pred, dtype=None)
    eq = (yt.local.double() == yp.local.double()).double()
    t = torch.stack([eq.sum(),
                     torch.tensor(float(eq.numel()), dtype=torch.float64,
                                  device=eq.device)])
    if yt.comm.world_size > 1:
        yt.comm.all_reduce(t)
    n = float(t[1])
    return float(t[0]) / max(n, 1.0)


def log_loss(y_true, y_prob, eps: float = 1e-15) -> float:
    """Binary or multiclass log loss. y_prob: [n] (binary p of class 1)
    or [n, k] probabilities."""
    yt = check_array(y_true, dtype=None)
    yp = check_array(y_prob, dtype=None)
    p = yp.local.double().clamp(eps, 1 - eps)
    t = yt.local
    if p.dim() == 1:
        tl = t.double()
        ll = -(tl * p.log() + (1 - tl) * (1 - p).log()).sum()
        n = float(p.shape[0])
    else:
        # NOTE: no row renormalization — the installed sklearn (1.7 era)
        # clips and scores y_prob as given (renormalization was removed);
        # verified numerically against sklearn.metrics.log_loss
        idx = t.long()
        ll = -(p[torch.arange(p.shape[0], device=p.device), idx].log()).sum()
        n = float(p.shape[0])
    buf = torch.stack([ll, torch.tensor(n, dtype=torch.float64,
                                        device=p.device)])
    if yt.comm.world_size > 1:
        yt.comm.all_reduce(buf)
    return float(buf[0]) / max(float(buf[1]), 1.0)
