"""SpectralClustering via Nyström approximation
(dask_ml/cluster/spectral.py parity — SURVEY §2.1).

Sample n_components rows globally, exact kernel on the sample, blockwise
kernel between the sample and every shard (map_blocks analog), orthogonalize
through the sample-kernel inverse square root, embed, row-normalize, and
cluster the embedding with the distributed KMeans.
"""

from __future__ import annotations

from typing import Optional, Union

import torch

from ..base import BaseEstimator, check_array, check_is_fitted
from ..core.array import DistArray
from ..core import random as mlrand
from .k_means import KMeans


def _rbf(a: torch.Tensor, b: torch.Tensor, gamma: float) -> torch.Tensor:
    an = (a * a).sum(dim=1)
    bn = (b * b).sum(dim=1)
    d2 = (an[:, None] + bn[None, :] - 2.0 * (a @ b.t())).clamp(min=0)
    return torch.exp(-gamma * d2)


class SpectralClustering(BaseEstimator):
    def __init__(self, n_clusters: int = 8, n_components: int = 100,
                 gamma: Union[float, None] = None, affinity: str = "rbf",
                 random_state: int = 0, kmeans_params: Optional[dict] = None,
                 persist_embedding: bool = False):
        self.n_clusters = n_clusters
        self.n_components = n_components
        self.gamma = gamma
        self.affinity = affinity
        self.random_state = random_state
        self.kmeans_params = kmeans_params
        self.persist_embedding = persist_embedding

    def fit(self, X, y=None):
        X = check_array(X)
        comm = X.comm
        d = X.ncols
        n = X.global_rows
        s = min(self.n_components, n)
        gamma = self.gamma if self.gamma is not None else 1.0 / d
        # sample s global rows (replicated deterministic choice)
        g = mlrand.generator(int(self.random_state or 0), shard_id=0x5bec,
                             device="cpu")
        idx = torch.randperm(n, generator=g)[:s].sort().values
        # gather sampled rows to every rank (tiny: one allreduce of [s, d])
        lo = X.row_offset
        hi = lo + X.local.shape[0]
        sample = torch.zeros(s, d, dtype=torch.float64, device=X.device)
        mask = (idx >= lo) & (idx < hi)
        own = idx[mask]
        if own.numel():
            sample[mask.nonzero()[:, 0]] = X.local[(own - lo)].double()
        if comm.world_size > 1:
            comm.all_reduce(sample)
        sample = sample.to(X.dtype)
        if self.affinity != "rbf":
            raise ValueError("only affinity='rbf' is implemented")
        A = _rbf(sample, sample, gamma).double()
        # A^{-1/2} via eigendecomposition (symmetric PSD)
        evals, evecs = torch.linalg.eigh(A)
        evals = evals.clamp(min=1e-8)
        Ainv_half = (evecs * (evals ** -0.5)[None, :]) @ evecs.t()
        # embedding rows per shard: K(x, sample) @ A^{-1/2}[:, :k]
        k = self.n_clusters
        proj = (Ainv_half @ evecs[:, -k:]).to(X.dtype)

        def _embed(t: torch.Tensor) -> torch.Tensor:
            B = _rbf(t, sample.to(t.device), gamma)
            E = B @ proj.to(t.device)
            return E / E.norm(dim=1, keepdim=True).clamp(min=1e-12)

        emb = X.map_local(_embed)
        km = KMeans(n_clusters=k, random_state=self.random_state,
                    **(self.kmeans_params or {}))
        km.fit(emb)
        self.labels_ = km.labels_
        self.assign_labels_ = km
        self._sample_ = sample
        self._proj_ = proj
        self._gamma_ = gamma
        self.n_features_in_ = d
        return self

    def fit_predict(self, X, y=None):
        return self.fit(X).labels_
