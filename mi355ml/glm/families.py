"""GLM families (dask_glm/families.py parity — SURVEY §2.2).

Each family names the fused-kernel loss (ops/glm.py) and provides the link
inverse (for predict) and the Hessian row-weights (for the newton solver).
"""

from __future__ import annotations

import torch


class Family:
    name: str = ""

    @staticmethod
    def predict(z: torch.Tensor) -> torch.Tensor:
        raise NotImplementedError

    @staticmethod
    def hessian_weights(z: torch.Tensor) -> torch.Tensor:
        raise NotImplementedError


class Logistic(Family):
    name = "logistic"

    @staticmethod
    def predict(z):
        return torch.sigmoid(z)

    @staticmethod
    def hessian_weights(z):
        s = torch.sigmoid(z)
        return s * (1.0 - s)


class Normal(Family):
    name = "normal"

    @staticmethod
    def predict(z):
        return z

    @staticmethod
    def hessian_weights(z):
        return torch.ones_like(z)


class Poisson(Family):
    name = "poisson"

    @staticmethod
    def predict(z):
        return torch.exp(z)

    @staticmethod
    def hessian_weights(z):
        return torch.exp(z)


FAMILIES = {"logistic": Logistic, "normal": Normal, "poisson": Poisson}


def get_family(name_or_cls) -> type:
    if isinstance(name_or_cls, str):
        return FAMILIES[name_or_cls]
    return name_or_cls
