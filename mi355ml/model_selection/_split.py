"""Splitters (dask_ml/model_selection/_split.py parity).

train_test_split / ShuffleSplit: per-shard Bernoulli masks from the
deterministic per-shard RNG streams (the reference's blockwise random-mask
pattern). KFold: contiguous global row ranges intersected with
each shard. Splits never move rows between ranks.
"""

from __future__ import annotations

from typing import Iterator, Optional, Tuple

import torch

from ..base import check_array
from ..core.array import DistArray, from_local
from ..core import random as mlrand


def _mask_split(X: DistArray, mask: torch.Tensor):
    tr = X.local[~mask]
    te = X.local[mask]
    return (from_local(tr.contiguous(), X.comm),
            from_local(te.contiguous(), X.comm))


def train_test_split(*arrays, test_size: float = None, train_size: float = None,
                     random_state: int = 0, shuffle: bool = True):
    """Split each array into (train, test) with consistent per-row masks."""
    if test_size is None:
        test_size = 0.25 if train_size is None else 1.0 - train_size
    arrays = [check_array(a, dtype=None) for a in arrays]
    X0 = arrays[0]
    if shuffle:
        g = mlrand.generator(int(random_state or 0), shard_id=X0.comm.rank,
                             stream=13, device=X0.device)
        u = torch.rand(X0.local.shape[0], generator=g, device=X0.device)
        mask = u < test_size
    else:
        # contiguous split (reference shuffle=False: trailing rows = test)
        lo = X0.row_offset
        idx = torch.arange(lo, lo + X0.local.shape[0], device=X0.device)
        cut = int(round(X0.global_rows * (1.0 - float(test_size))))
        mask = idx >= cut
    out = []
    for a in arrays:
        if a.local.shape[0] != X0.local.shape[0]:
            raise ValueError("arrays have inconsistent local lengths")
        tr, te = _mask_split(a, mask)
        out.extend([tr, te])
    return out


class ShuffleSplit:
    def __init__(self, n_splits: int = 10, test_size: float = 0.1,
                 train_size=None, random_state: int = 0):
        self.n_splits = n_splits
        self.test_size = test_size
        self.train_size = train_size
        self.random_state = random_state

    def get_n_splits(self, X=None, y=None, groups=None):
        return self.n_splits

    def split(self, X, y=None, groups=None) -> Iterator[Tuple]:
        X = check_array(X, dtype=None)
        for i in range(self.n_splits):
            g = mlrand.generator(int(self.random_state or 0) + i,
                                 shard_id=X.comm.rank, stream=17,
                                 device=X.device)
            u = torch.rand(X.local.shape[0], generator=g, device=X.device)
            yield u >= self.test_size, u < self.test_size  # local bool masks


class KFold:
    """Contiguous global folds (reference KFold splits whole blocks)."""

    def __init__(self, n_splits: int = 5, shuffle: bool = False,
                 random_state: int = 0):
        self.n_splits = n_splits
        self.shuffle = shuffle
        self.random_state = random_state

    def get_n_splits(self, X=None, y=None, groups=None):
        return self.n_splits

    def split(self, X, y=None, groups=None) -> Iterator[Tuple]:
        X = check_array(X, dtype=None)
        n = X.global_rows
        lo = X.row_offset
        m = X.local.shape[0]
        if self.shuffle:
            # rows keep their rank; fold membership is a deterministic
            # per-row random draw (the no-row-movement analog of the
            # reference's shuffled block folds; folds are equal-sized in
            # expectation)
            g = mlrand.generator(int(self.random_state or 0),
                                 shard_id=X.comm.rank, stream=29,
                                 device=X.device)
            fold = torch.randint(0, self.n_splits, (m,), generator=g,
                                 device=X.device)
            for f in range(self.n_splits):
                te = fold == f
                yield ~te, te
            return
        idx = torch.arange(lo, lo + m, device=X.device)
        base, rem = divmod(n, self.n_splits)
        start = 0
        for f in range(self.n_splits):
            size = base + (1 if f < rem else 0)
            te = (idx >= start) & (idx < start + size)
            yield ~te, te
            start += size


def check_cv(cv=None):
    if cv is None:
        return KFold(n_splits=5)
    if isinstance(cv, int):
        return KFold(n_splits=cv)
    return cv


def mask_take(arrays, mask: torch.Tensor):
    """Apply a local boolean mask to a list of DistArrays -> new DistArrays."""
    return [from_local(a.local[mask].contiguous(), a.comm) for a in arrays]
