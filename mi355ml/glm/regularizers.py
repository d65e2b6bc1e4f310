"""Regularizers with proximal operators (dask_glm/regularizers.py parity).

K17 in SURVEY §2.7: the prox ops are tiny F-length device ops; plain torch is
already optimal (latency-bound, not a HIP-kernel candidate).
"""

from __future__ import annotations

import torch


class Regularizer:
    def f(self, w: torch.Tensor) -> float:
        raise NotImplementedError

    def gradient(self, w: torch.Tensor) -> torch.Tensor:
        raise NotImplementedError

    def proximal_operator(self, w: torch.Tensor, t: float) -> torch.Tensor:
        raise NotImplementedError


class L2(Regularizer):
    name = "l2"

    def f(self, w):
        return 0.5 * float((w.double() * w.double()).sum())

    def gradient(self, w):
        return w.double()

    def proximal_operator(self, w, t):
        return w / (1.0 + t)

    def hessian_diag(self, w):
        return torch.ones_like(w, dtype=torch.float64)


class L1(Regularizer):
    name = "l1"

    def f(self, w):
        return float(w.double().abs().sum())

    def gradient(self, w):
        # subgradient (used only for convergence diagnostics)
        return torch.sign(w).double()

    def proximal_operator(self, w, t):
        return torch.sign(w) * (w.abs() - t).clamp(min=0.0)


class ElasticNet(Regularizer):
    name = "elastic_net"

    def __init__(self, weight: float = 0.5):
        self.weight = weight  # fraction of L1

    def f(self, w):
        a = self.weight
        return a * L1().f(w) + (1 - a) * L2().f(w)

    def gradient(self, w):
        a = self.weight
        return a * L1().gradient(w) + (1 - a) * L2().gradient(w)

    def proximal_operator(self, w, t):
        a = self.weight
        soft = torch.sign(w) * (w.abs() - t * a).clamp(min=0.0)
        return soft / (1.0 + t * (1 - a))


REGULARIZERS = {"l1": L1, "l2": L2, "elastic_net": ElasticNet}


def get_regularizer(name_or_obj) -> Regularizer:
    if isinstance(name_or_obj, Regularizer):
        return name_or_obj
    return REGULARIZERS[name_or_obj]()
