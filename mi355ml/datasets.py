"""Synthetic dataset generators (dask_ml/datasets.py parity).

Per-shard generation with a SHARED projection drawn from the master seed and
independent per-shard noise streams (the reference's per-block-seed pattern,
dask/array/random.py:75): each rank generates only its own rows, directly on
its device — no scatter/broadcast of data ("no broadcast needed if
each rank generates its own shard").
"""

from __future__ import annotations

from typing import Tuple, Union

import torch

from .core.array import DistArray
from .core import random as mlrand
from .parallel import comm as comm_mod


def _master_gen(seed: int, device) -> torch.Generator:
    return mlrand.generator(seed, shard_id=0xFFFF, stream=0, device=device)


def _shard_gen(seed: int, comm, stream: int = 1) -> torch.Generator:
    return mlrand.generator(seed, shard_id=comm.rank, stream=stream,
                            device=comm.device)


def _shard_rows(n_samples: int, comm) -> Tuple[int, int]:
    lo, hi = comm.shard_bounds(n_samples)
    return lo, hi


def make_blobs(n_samples: int = 100, n_features: int = 2,
               centers: Union[int, torch.Tensor] = 3,
               cluster_std: float = 1.0, center_box: Tuple[float, float] = (-10.0, 10.0),
               random_state: int = 0, comm=None,
               dtype: torch.dtype = torch.float32):
    comm = comm or comm_mod.default()
    dev = comm.device
    mg = _master_gen(random_state, dev)
    if isinstance(centers, int):
        k = centers
        c = (torch.rand(k, n_features, generator=mg, device=dev, dtype=dtype)
             * (center_box[1] - center_box[0]) + center_box[0])
    else:
        c = torch.as_tensor(centers, dtype=dtype, device=dev)
        k = c.shape[0]
    lo, hi = _shard_rows(n_samples, comm)
    m = hi - lo
    sg = _shard_gen(random_state, comm)
    labels = torch.randint(0, k, (m,), generator=sg, device=dev)
    X = (torch.randn(m, n_features, generator=sg, device=dev, dtype=dtype)
         * cluster_std + c[labels])
    return (DistArray(X, n_samples, lo, comm),
            DistArray(labels.to(torch.int64), n_samples, lo, comm))


def make_classification(n_samples: int = 100, n_features: int = 20,
                        n_informative: int = 2, scale: float = 1.0,
                        flip_y: float = 0.01, random_state: int = 0,
                        comm=None, dtype: torch.dtype = torch.float32):
    """Binary classification: X ~ N(0, 1); y = 1[sigmoid(X·w) > u] with a
    shared sparse informative weight vector w."""
    comm = comm or comm_mod.default()
    dev = comm.device
    mg = _master_gen(random_state, dev)
    w = torch.zeros(n_features, device=dev, dtype=dtype)
    idx = torch.randperm(n_features, generator=mg, device=dev)[:n_informative]
    w[idx] = torch.randn(n_informative, generator=mg, device=dev, dtype=dtype) * 2.0
    lo, hi = _shard_rows(n_samples, comm)
    m = hi - lo
    sg = _shard_gen(random_state, comm)
    X = torch.randn(m, n_features, generator=sg, device=dev, dtype=dtype) * scale
    logits = X @ w
    p = torch.sigmoid(logits)
    u = torch.rand(m, generator=sg, device=dev, dtype=dtype)
    y = (p > u).to(dtype)
    if flip_y > 0:
        flip = torch.rand(m, generator=sg, device=dev, dtype=dtype) < flip_y
        y = torch.where(flip, 1.0 - y, y)
    return (DistArray(X, n_samples, lo, comm),
            DistArray(y, n_samples, lo, comm))


def make_regression(n_samples: int = 100, n_features: int = 20,
                    n_informative: int = 10, noise: float = 0.0,
                    coef: bool = False, random_state: int = 0, comm=None,
                    dtype: torch.dtype = torch.float32):
    comm = comm or comm_mod.default()
    dev = comm.device
    mg = _master_gen(random_state, dev)
    w = torch.zeros(n_features, device=dev, dtype=dtype)
    idx = torch.randperm(n_features, generator=mg, device=dev)[:n_informative]
    w[idx] = 100.0 * torch.rand(n_informative, generator=mg, device=dev,
                                dtype=dtype)
    lo, hi = _shard_rows(n_samples, comm)
    m = hi - lo
    sg = _shard_gen(random_state, comm)
    X = torch.randn(m, n_features, generator=sg, device=dev, dtype=dtype)
    y = X @ w
    if noise > 0:
        y = y + torch.randn(m, generator=sg, device=dev, dtype=dtype) * noise
    Xd = DistArray(X, n_samples, lo, comm)
    yd = DistArray(y, n_samples, lo, comm)
    if coef:
        return Xd, yd, w
    return Xd, yd


def make_counts(n_samples: int = 100, n_features: int = 20,
                n_informative: int = 10, scale: float = 1.0,
                random_state: int = 0, comm=None,
                dtype: torch.dtype = torch.float32):
    """Poisson-target data: y ~ Poisson(exp(X·w · scale))."""
    comm = comm or comm_mod.default()
    dev = comm.device
    mg = _master_gen(random_state, dev)
    w = torch.zeros(n_features, device=dev, dtype=dtype)
    idx = torch.randperm(n_features, generator=mg, device=dev)[:n_informative]
    w[idx] = torch.randn(n_informative, generator=mg, device=dev, dtype=dtype) * 0.3
    lo, hi = _shard_rows(n_samples, comm)
    m = hi - lo
    sg = _shard_gen(random_state, comm)
    X = torch.randn(m, n_features, generator=sg, device=dev, dtype=dtype) * scale
    rate = torch.exp((X @ w).clamp(max=20.0))
    y = torch.poisson(rate, generator=sg).to(dtype)
    return (DistArray(X, n_samples, lo, comm),
            DistArray(y, n_samples, lo, comm))
