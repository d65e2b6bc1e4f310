"""Solver-state checkpoint/resume (SURVEY §5.3/§5.4 rebuild).

The reference recovers from worker death by lineage re-execution; at 8
ranks on one node the cheaper design is estimator-level checkpoints —
solvers are iterative, and their full state is tiny ((β, iter) for GLMs,
(centers, iter) for KMeans). save_checkpoint moves device tensors to host
(rank 0 writes, all ranks barrier); after a rank respawn, load_checkpoint
restores the replicated state and fitting resumes via warm-start
(`KMeans(init=centers)`, solver `w0`).

Model persistence itself stays joblib-compatible (BaseEstimator.__getstate__
moves tensors to host numpy).
"""

from __future__ import annotations

import os
import tempfile
from typing import Any, Dict, Optional

import torch

from .parallel import comm as comm_mod


def save_checkpoint(path: str, state: Dict[str, Any],
                    comm=None) -> None:
    """Atomically write a host-side checkpoint from rank 0."""
    comm = comm or comm_mod.default()
    if comm.rank == 0:
        host_state = {
            k: (v.detach().cpu() if isinstance(v, torch.Tensor) else v)
            for k, v in state.items()}
        d = os.path.dirname(os.path.abspath(path)) or "."
        os.makedirs(d, exist_ok=True)
        fd, tmp = tempfile.mkstemp(dir=d, suffix=".ckpt.tmp")
        os.close(fd)
        torch.save(host_state, tmp)
        os.replace(tmp, path)
    comm.barrier()


def load_checkpoint(path: str, device=None, comm=None
                    ) -> Optional[Dict[str, Any]]:
    """Load a checkpoint onto every rank's device; None if absent."""
    comm = comm or comm_mod.default()
    if not os.path.exists(path):
        return None
    state = torch.load(path, map_location="cpu", weights_only=False)
    dev = device if device is not None else comm.device
    return {k: (v.to(dev) if isinstance(v, torch.Tensor) else v)
            for k, v in state.items()}


class CheckpointManager:
    """Periodic checkpointing for iterative solvers.

        mgr = CheckpointManager("km.ckpt", every=10)
        for it in ...:
            ...
            mgr.step(it, centers=centers)
    """

    def __init__(self, path: Optional[str], every: int = 10, comm=None):
        self.path = path
        self.every = every
        self.comm = comm or comm_mod.default()

    def step(self, iteration: int, **state) -> None:
        if self.path and self.every and iteration % self.every == 0:
            save_checkpoint(self.path, {"iteration": iteration, **state},
                            self.comm)

    def resume(self) -> Optional[Dict[str, Any]]:
        if not self.path:
            return None
        return load_checkpoint(self.path, comm=self.comm)
