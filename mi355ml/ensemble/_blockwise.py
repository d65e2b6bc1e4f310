"""Blockwise voting ensembles (dask_ml/ensemble/_blockwise.py parity).

fit: one clone per row block, embarrassingly parallel — each rank fits
clones on its local slabs, then ONE all_gather_object replicates the
(small) fitted estimators everywhere. predict: majority vote / mean across
estimators_, blockwise on each shard.
"""

from __future__ import annotations

from typing import List

import numpy as np
import torch
from sklearn.base import clone

from .. import config
from ..base import BaseEstimator, check_array, check_is_fitted
from ..core.array import DistArray
from ..metrics import accuracy_score, r2_score


class _BlockwiseBase(BaseEstimator):
    def __init__(self, estimator):
        self.estimator = estimator

    def _fit(self, X: DistArray, y: DistArray, **kwargs):
        bs = int(config.get("array.block_rows"))
        x = X.local
        yl = y.local
        blocks = []
        for lo in range(0, max(x.shape[0], 1), bs):
            xb = x[lo:lo + bs]
            if xb.shape[0]:
                blocks.append((xb.cpu().numpy(),
                               yl[lo:lo + bs].cpu().numpy()))

        def _fit_block(args):
            xb, yb = args
            est = clone(self.estimator)
            est.fit(xb, yb, **kwargs)
            return est

        # block fits are independent sklearn CPU fits — fan out over an
        # intra-rank thread pool (same knob as the adaptive searches)
        nthreads = int(config.get("search.threads") or 0)
        if nthreads > 1 and len(blocks) > 1:
            from concurrent.futures import ThreadPoolExecutor
            ovs = config.capture_overlays()

            def _fit_w(b):
                config.install_overlays(ovs)
                return _fit_block(b)

            with ThreadPoolExecutor(min(nthreads, len(blocks))) as pool:
                local_models = list(pool.map(_fit_w, blocks))
        else:
            local_models = [_fit_block(b) for b in blocks]
        gathered = X.comm.all_gather_object(local_models)
        self.estimators_ = [m for part in gathered for m in part]
        self.n_features_in_ = X.ncols
        return self

    def fit(self, X, y, **kwargs):
        X = check_array(X, dtype=None)
        y = check_array(y, dtype=None)
        return self._fit(X, y, **kwargs)


class BlockwiseVotingClassifier(_BlockwiseBase):
    def fit(self, X, y, classes=None, **kwargs):
        super().fit(X, y, **kwargs)
        if classes is not None:
            self.classes_ = np.asarray(classes)
        else:
            cs = set()
            for est in self.estimators_:
                cs.update(np.asarray(est.classes_).tolist())
            self.classes_ = np.asarray(sorted(cs))
        return self

    def predict(self, X):
        check_is_fitted(self, "estimators_")
        X = check_array(X, dtype=None)
        classes = self.classes_
        ests = self.estimators_

        def _vote(t: torch.Tensor) -> torch.Tensor:
            xn = t.cpu().numpy()
            votes = np.zeros((xn.shape[0], len(classes)), dtype=np.int64)
            lut = {c: i for i, c in enumerate(classes.tolist())}
            for est in ests:
                p = est.predict(xn)
                idx = np.vectorize(lut.__getitem__)(p)
                votes[np.arange(xn.shape[0]), idx] += 1
            out = classes[votes.argmax(axis=1)]
            return torch.as_tensor(out, device=t.device)

        return X.map_local(_vote)

    def predict_proba(self, X):
        check_is_fitted(self, "estimators_")
        X = check_array(X, dtype=None)
        ests = self.estimators_

        def _avg(t: torch.Tensor) -> torch.Tensor:
            xn = t.cpu().numpy()
            acc = None
            for est in ests:
                p = est.predict_proba(xn)
                acc = p if acc is None else acc + p
            return torch.as_tensor(acc / len(ests), device=t.device)

        return X.map_local(_avg)

    def score(self, X, y):
        return accuracy_score(check_array(y, dtype=None), self.predict(X))


class BlockwiseVotingRegressor(_BlockwiseBase):
    def predict(self, X):
        check_is_fitted(self, "estimators_")
        X = check_array(X, dtype=None)
        ests = self.estimators_

        def _mean(t: torch.Tensor) -> torch.Tensor:
            xn = t.cpu().numpy()
            acc = None
            for est in ests:
                p = est.predict(xn)
                acc = p if acc is None else acc + p
            return torch.as_tensor(acc / len(ests), device=t.device)

        return X.map_local(_mean)

    def score(self, X, y):
        return r2_score(check_array(y, dtype=None), self.predict(X))
