"""Estimator base classes: sklearn API contract + joblib persistence.

Estimators follow the sklearn contract the reference mirrors
(sklearn/base.py:141,178,202 get_params/set_params; :668 fit_transform;
:470 score) by subclassing sklearn.base.BaseEstimator — same choice as
dask-ml. Fitted attributes live as small replicated torch tensors on the
rank's device; `__getstate__` moves them to host numpy so `joblib.dump`
produces portable files (SURVEY §2.1 model-persistence contract).
"""

from __future__ import annotations

from typing import Union

import numpy as np
import torch
from sklearn.base import BaseEstimator as _SkBase
from sklearn.base import ClassifierMixin, RegressorMixin, TransformerMixin  # noqa: F401

from .core.array import DistArray, from_numpy, shard_rows
from .parallel import comm as comm_mod

ArrayLike = Union[DistArray, np.ndarray, torch.Tensor]


def check_array(X: ArrayLike, dtype: torch.dtype = torch.float32,
                comm=None, col_ok: bool = False) -> DistArray:
    """Coerce input to a row-sharded DistArray (dask-tolerant check_array
    analog, dask_ml/utils.py). Column-sharded (ColDistArray) inputs pass
    through ONLY when the caller opts in with col_ok=True — estimators
    that assume row shards would silently misread a column slice as a
    row shard otherwise."""
    if isinstance(X, DistArray):
        if dtype is not None and X.dtype != dtype:
            from .core.spill import SpilledDistArray
            if isinstance(X, SpilledDistArray):
                return SpilledDistArray(X.local.to(dtype), X.global_rows,
                                        X.row_offset, X.comm, X.block_rows)
            return X.astype(dtype)
        return X
    from .core.colarray import ColDistArray, Replicated
    if isinstance(X, ColDistArray):
        if not col_ok:
            raise TypeError(
                "column-sharded (ColDistArray) input is only supported by "
                "the feature-parallel paths (GLM lbfgs/GD/prox, "
                "TruncatedSVD, PCA); this estimator needs row-sharded "
                "data — use mi355ml.from_numpy / shard_rows")
        # column-sharded (wide-X) arrays pass through untouched
        if dtype is not None and X.dtype != dtype:
            return X.astype(dtype)
        return X
    if isinstance(X, Replicated):
        # replicated per-row result of a feature-parallel op: every rank
        # holds the identical full tensor — slice it into a valid
        # row-sharded DistArray
        t = X.local
        if dtype is not None and t.dtype != dtype:
            t = t.to(dtype)
        return shard_rows(t, comm or X.comm)
    if isinstance(X, np.ndarray):
        t = torch.from_numpy(np.ascontiguousarray(X))
    elif isinstance(X, torch.Tensor):
        t = X
    else:
        t = torch.as_tensor(np.asarray(X))
    if dtype is not None and t.dtype != dtype:
        t = t.to(dtype)
    return shard_rows(t, comm)


def check_X_y(X: ArrayLike, y: ArrayLike, dtype: torch.dtype = torch.float32,
              y_dtype: torch.dtype = torch.float32, comm=None,
              col_ok: bool = False):
    Xd = check_array(X, dtype, comm, col_ok=col_ok)
    from .core.colarray import ColDistArray, Replicated
    if isinstance(Xd, ColDistArray):
        # col-sharded X holds ALL rows on every rank: y must be the full
        # replicated label vector, not a row shard
        if isinstance(y, DistArray):
            yt = y.gather()
        elif isinstance(y, Replicated):
            yt = y.local
        elif isinstance(y, torch.Tensor):
            yt = y
        else:
            yt = torch.as_tensor(np.asarray(y))
        yt = yt.to(device=Xd.device, dtype=y_dtype or yt.dtype)
        if yt.shape[0] != Xd.global_rows:
            raise ValueError(
                f"X and y have inconsistent lengths: {Xd.global_rows} vs "
                f"{yt.shape[0]}")
        return Xd, Replicated(yt, Xd.comm)
    yd = check_array(y, y_dtype, Xd.comm)
    if Xd.global_rows != yd.global_rows:
        raise ValueError(
            f"X and y have inconsistent lengths: {Xd.global_rows} vs "
            f"{yd.global_rows}")
    return Xd, yd


def check_is_fitted(est, attr: str) -> None:
    if not hasattr(est, attr):
        raise RuntimeError(
            f"{type(est).__name__} is not fitted; call fit() first.")


def _to_host(v):
    if isinstance(v, torch.Tensor):
        return ("__mi355ml_tensor__", v.detach().cpu().numpy())
    if isinstance(v, DistArray):
        # persist the local shard + layout; restored against the current comm
        return ("__mi355ml_distarray__", v.local.detach().cpu().numpy(),
                v.global_rows, v.row_offset)
    return v


def _from_host(v, device):
    if isinstance(v, tuple) and len(v) >= 2 and v[0] == "__mi355ml_tensor__":
        return torch.from_numpy(np.ascontiguousarray(v[1])).to(device)
    if isinstance(v, tuple) and len(v) == 4 and v[0] == "__mi355ml_distarray__":
        local = torch.from_numpy(np.ascontiguousarray(v[1])).to(device)
        return DistArray(local, v[2], v[3])
    return v


class BaseEstimator(_SkBase):
    """sklearn-compatible base with device-aware pickling."""

    def __getstate__(self):
        state = self.__dict__.copy()
        return {k: _to_host(v) for k, v in state.items()}

    def __setstate__(self, state):
        device = comm_mod.default().device
        self.__dict__.update(
            {k: _from_host(v, device) for k, v in state.items()})

    # convenience for subclasses
    def _device(self) -> torch.device:
        return comm_mod.default().device


def as_result(t: torch.Tensor, like: ArrayLike):
    """Return small fitted/derived results in a caller-friendly form:
    numpy in, numpy out; tensors pass through."""
    if isinstance(like, np.ndarray):
        return t.detach().cpu().numpy()
    return t
