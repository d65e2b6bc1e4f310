"""ParallelPostFit / Incremental (dask_ml/wrappers.py + _partial.py parity).

ParallelPostFit: fit a (sklearn) estimator on gathered data once, then run
predict/transform/predict_proba blockwise on each rank's shard — the
reference's map_blocks-with-broadcast-model pattern (C10), except the model
is already replicated because every rank runs the same fit.

Incremental: the reference's sequential partial_fit chain through blocks
(model object threaded block to block). MI355X mapping: ranks run their
local blocks in rank order, handing the pickled model to the next rank via
one tiny object broadcast (same pattern as IncrementalPCA).
"""

from __future__ import annotations

import numpy as np
import torch
from sklearn.base import clone

from .base import BaseEstimator, check_array
from .core.array import DistArray
from .core.bag import DistList
from .core import random as mlrand
from .metrics import accuracy_score, r2_score


def _to_numpy(t: torch.Tensor) -> np.ndarray:
    return t.detach().cpu().numpy()


def _is_partitioned(X) -> bool:
    """DistList of scipy-sparse/numpy partitions (text-pipeline outputs:
    the reference's array-of-sparse-blocks with unknown row chunks)."""
    return isinstance(X, DistList)


def _blockwise_method(est, name, X: DistArray, out_dtype=torch.float32):
    fn = getattr(est, name)

    def _apply(t: torch.Tensor) -> torch.Tensor:
        res = fn(_to_numpy(t))
        return torch.as_tensor(np.ascontiguousarray(res),
                               device=t.device)

    return X.map_local(_apply)


class ParallelPostFit(BaseEstimator):
    def __init__(self, estimator=None, scoring=None):
        self.estimator = estimator
        self.scoring = scoring

    def fit(self, X, y=None, **kwargs):
        est = clone(self.estimator)
        if _is_partitioned(X):
            import scipy.sparse as sp
            parts = X.gather()
            Xc = sp.vstack(parts) if parts and sp.issparse(parts[0]) \
                else np.concatenate(parts)
            yc = None
            if y is not None:
                yg = (y.gather() if _is_partitioned(y)
                      else check_array(y, dtype=None).gather())
                yc = (np.concatenate(yg) if isinstance(yg, list)
                      else _to_numpy(yg))
            est.fit(Xc, yc, **kwargs) if yc is not None else est.fit(
                Xc, **kwargs)
            self.estimator_ = est
            return self
        X = check_array(X, dtype=None)
        Xc = _to_numpy(X.gather())
        if y is not None:
            y = check_array(y, dtype=None)
            est.fit(Xc, _to_numpy(y.gather()), **kwargs)
        else:
            est.fit(Xc, **kwargs)
        self.estimator_ = est
        return self

    @property
    def classes_(self):
        return self.estimator_.classes_

    def predict(self, X):
        if _is_partitioned(X):
            fn = getattr(self.estimator_, "predict")
            return X.map(fn)
        X = check_array(X, dtype=None)
        return _blockwise_method(self.estimator_, "predict", X)

    def predict_proba(self, X):
        if _is_partitioned(X):
            fn = getattr(self.estimator_, "predict_proba")
            return X.map(fn)
        X = check_array(X, dtype=None)
        return _blockwise_method(self.estimator_, "predict_proba", X)

    def predict_log_proba(self, X):
        if _is_partitioned(X):
            fn = getattr(self.estimator_, "predict_log_proba")
            return X.map(fn)
        X = check_array(X, dtype=None)
        return _blockwise_method(self.estimator_, "predict_log_proba", X)

    def transform(self, X):
        if _is_partitioned(X):
            fn = getattr(self.estimator_, "transform")
            return X.map(fn)
        X = check_array(X, dtype=None)
        return _blockwise_method(self.estimator_, "transform", X)

    def score(self, X, y):
        yp = self.predict(X)
        y = check_array(y, dtype=None)
        if hasattr(self.estimator_, "_estimator_type") and \
                self.estimator_._estimator_type == "regressor":
            return r2_score(y, yp)
        return accuracy_score(y, yp)


class Incremental(ParallelPostFit):
    """Sequential partial_fit through all blocks (+ optional per-epoch block
    shuffle), sklearn-estimator semantics preserved exactly."""

    def __init__(self, estimator=None, scoring=None,
                 shuffle_blocks: bool = True, random_state: int = 0,
                 assume_equal_chunks: bool = True):
        super().__init__(estimator, scoring)
        self.shuffle_blocks = shuffle_blocks
        self.random_state = random_state
        self.assume_equal_chunks = assume_equal_chunks

    def _partial_fit_pass(self, est, X, y, classes, epoch: int,
                          **kwargs):
        comm = X.comm
        order = list(range(comm.world_size))
        if self.shuffle_blocks and comm.world_size > 1:
            import random as _pyrand
            rng = _pyrand.Random(
                mlrand.shard_seed(int(self.random_state or 0), epoch, 23))
            rng.shuffle(order)
        for r in order:
            if comm.rank == r and est is not None:
                if _is_partitioned(X):
                    import scipy.sparse as sp
                    parts = X.local
                    if y is not None and _is_partitioned(y) and \
                            len(y.local) != len(parts):
                        raise ValueError(
                            f"X has {len(parts)} local partitions but y has "
                            f"{len(y.local)}; X and y must be partitioned "
                            "identically (same DistList split)")
                    xb = (sp.vstack(parts) if parts and
                          sp.issparse(parts[0]) else
                          (np.concatenate(parts) if parts else
                           np.zeros((0, 0))))
                    yb = (np.concatenate(y.local) if _is_partitioned(y)
                          else _to_numpy(y.local)) if y is not None else None
                else:
                    xb = _to_numpy(X.local if X.ndim == 2
                                   else X.local[:, None])
                    yb = _to_numpy(y.local) if y is not None else None
                if yb is not None and xb.shape[0] != len(yb):
                    raise ValueError(
                        f"local X block has {xb.shape[0]} rows but y has "
                        f"{len(yb)}; X and y must shard identically")
                if xb.shape[0]:
                    if classes is not None and yb is not None:
                        # sklearn accepts (identical) classes on every call
                        est.partial_fit(xb, yb, classes=classes, **kwargs)
                    elif yb is not None:
                        est.partial_fit(xb, yb, **kwargs)
                    else:
                        est.partial_fit(xb, **kwargs)
            if comm.world_size > 1:
                est = comm.broadcast_object(est if comm.rank == r else None,
                                            src=r)
        return est

    def fit(self, X, y=None, classes=None, epochs: int = 1, **kwargs):
        if not _is_partitioned(X):
            X = check_array(X, dtype=None)
        if y is not None and not _is_partitioned(y):
            y = check_array(y, dtype=None)
        est = clone(self.estimator)
        if classes is None and y is not None and \
                hasattr(est, "partial_fit") and \
                est._estimator_type == "classifier":
            if _is_partitioned(y):
                vals = set()
                for part in y.comm.all_gather_object(
                        sorted(set(np.concatenate(y.local).tolist()
                                   if y.local else []))):
                    vals.update(part)
                classes = np.asarray(sorted(vals))
            else:
                from .ops.unique import unique
                classes = _to_numpy(unique(y))
        for ep in range(epochs):
            est = self._partial_fit_pass(est, X, y, classes, ep, **kwargs)
        self.estimator_ = est
        return self

    def partial_fit(self, X, y=None, classes=None, **kwargs):
        X = check_array(X, dtype=None)
        if y is not None:
            y = check_array(y, dtype=None)
        est = getattr(self, "estimator_", None)
        if est is None:
            est = clone(self.estimator)
        est = self._partial_fit_pass(est, X, y, classes, 0, **kwargs)
        self.estimator_ = est
        return self
