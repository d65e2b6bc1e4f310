"""Data-plane communicator: RCCL over xGMI (one process per GPU), gloo on CPU.

This replaces the reference's entire distributed runtime for dense math: every
cross-chunk tree-reduce the reference schedules as a task graph
(dask/array/reductions.py:200 `_tree_reduce`) is ONE collective here
(SURVEY §2.7 C-table). Control-plane (Hyperband trial orchestration) lives in
mi355ml/parallel/coordinator.py, not here.

Design:
  - SPMD, eager. `comm.default()` reads RANK/WORLD_SIZE/MASTER_* from the env
    (torchrun-compatible) and initializes torch.distributed once.
  - backend "nccl" IS RCCL on ROCm; "gloo" used for CPU multi-process tests.
  - world_size==1 (no env): collectives are no-ops / local identities, so the
    same estimator code runs single-process.
  - A dedicated side stream for collectives lets the hot solvers overlap the
    (tiny, latency-bound) per-iteration all-reduces with the next block's
    compute.
"""

from __future__ import annotations

import datetime
import os
from typing import List, Optional

import torch
import torch.distributed as dist

from .. import config

_DEFAULT: Optional["Communicator"] = None


class Communicator:
    """Thin SPMD wrapper over torch.distributed (RCCL on GPU, gloo on CPU)."""

    def __init__(self, rank: int, world_size: int, device: torch.device,
                 group: Optional[object] = None):
        self.rank = rank
        self.world_size = world_size
        self.device = device
        self.group = group
        self._comm_stream: Optional[torch.cuda.Stream] = None
        if device.type == "cuda" and config.get("comm.overlap"):
            self._comm_stream = torch.cuda.Stream(device=device)

    # -- collectives ------------------------------------------------------
    def _count(self, kind: str, t: torch.Tensor) -> None:
        """Per-collective call/byte counters (diagnostics.COUNTERS; shows
        up in serving /metrics — SURVEY §5.5 observability)."""
        from ..diagnostics import COUNTERS
        COUNTERS.inc(f"comm.{kind}.calls")
        COUNTERS.inc(f"comm.{kind}.bytes",
                     float(t.numel() * t.element_size()))

    def all_reduce(self, t: torch.Tensor, op: str = "sum") -> torch.Tensor:
        if self.world_size == 1:
            return t
        ops = {"sum": dist.ReduceOp.SUM, "max": dist.ReduceOp.MAX,
               "min": dist.ReduceOp.MIN}
        self._count("all_reduce", t)
        dist.all_reduce(t, op=ops[op], group=self.group)
        return t

    def all_reduce_async(self, t: torch.Tensor, op: str = "sum"):
        """Launch an all-reduce on the side stream; returns a handle with
        .wait() making the default stream depend on completion."""
        if self.world_size == 1:
            return _DoneHandle(t)
        if self._comm_stream is None:
            self.all_reduce(t, op)
            return _DoneHandle(t)
        cur = torch.cuda.current_stream(self.device)
        self._comm_stream.wait_stream(cur)
        with torch.cuda.stream(self._comm_stream):
            self.all_reduce(t, op)
        return _StreamHandle(t, self._comm_stream, self.device)

    def all_gather(self, t: torch.Tensor) -> List[torch.Tensor]:
        if self.world_size == 1:
            return [t]
        self._count("all_gather", t)
        out = [torch.empty_like(t) for _ in range(self.world_size)]
        dist.all_gather(out, t.contiguous(), group=self.group)
        return out

    def all_gather_varlen(self, t: torch.Tensor) -> List[torch.Tensor]:
        """All-gather of per-rank tensors with differing first-dim sizes
        (ragged: SURVEY C4/C12). Pads to max and slices back."""
        if self.world_size == 1:
            return [t]
        n = torch.tensor([t.shape[0]], dtype=torch.int64, device=t.device)
        counts = self.all_gather(n)
        counts = [int(c.item()) for c in counts]
        m = max(counts) if counts else 0
        pad_shape = (m,) + tuple(t.shape[1:])
        padded = torch.zeros(pad_shape, dtype=t.dtype, device=t.device)
        if t.shape[0]:
            padded[: t.shape[0]] = t
        parts = self.all_gather(padded)
        return [p[:c] for p, c in zip(parts, counts)]

    def broadcast(self, t: torch.Tensor, src: int = 0) -> torch.Tensor:
        if self.world_size == 1:
            return t
        self._count("broadcast", t)
        dist.broadcast(t, src=src, group=self.group)
        return t

    def barrier(self) -> None:
        if self.world_size > 1:
            dist.barrier(group=self.group)

    # -- control-plane object collectives (tiny, pickle-based) ------------
    def broadcast_object(self, obj, src: int = 0):
        if self.world_size == 1:
            return obj
        lst = [obj if self.rank == src else None]
        dist.broadcast_object_list(lst, src=src, group=self.group)
        return lst[0]

    def all_gather_object(self, obj) -> list:
        if self.world_size == 1:
            return [obj]
        out = [None] * self.world_size
        dist.all_gather_object(out, obj, group=self.group)
        return out

    # -- helpers ----------------------------------------------------------
    def shard_bounds(self, n_rows: int) -> tuple:
        """Contiguous row range [lo, hi) owned by this rank (balanced)."""
        base, rem = divmod(n_rows, self.world_size)
        lo = self.rank * base + min(self.rank, rem)
        hi = lo + base + (1 if self.rank < rem else 0)
        return lo, hi

    def __repr__(self) -> str:  # pragma: no cover
        return (f"Communicator(rank={self.rank}, world={self.world_size}, "
                f"device={self.device})")


class _DoneHandle:
    def __init__(self, t):
        self.tensor = t

    def wait(self):
        return self.tensor


class _StreamHandle:
    def __init__(self, t, stream, device):
        self.tensor = t
        self._stream = stream
        self._device = device

    def wait(self):
        torch.cuda.current_stream(self._device).wait_stream(self._stream)
        return self.tensor


def _elastic_store(rank: int, world: int, restart: int,
                   timeout) -> "dist.PrefixStore":
    """TCPStore + per-attempt prefix for torchelastic respawns. The master
    retries the bind: the previous attempt's rank-0 process may hold the
    port for a short window while it dies."""
    import time as _time
    host = os.environ.get("MASTER_ADDR", "127.0.0.1")
    port = int(os.environ.get("MASTER_PORT", "29531"))
    deadline = _time.monotonic() + timeout.total_seconds()
    last = None
    while True:
        try:
            ts = dist.TCPStore(host, port, world, rank == 0,
                               timeout=timeout)
            break
        except Exception as e:  # bind/connect race with the dying attempt
            last = e
            if _time.monotonic() > deadline:
                raise RuntimeError(
                    f"elastic store setup failed after retries: {last}")
            _time.sleep(0.25)
    return dist.PrefixStore(f"mi355ml/attempt-{restart}", ts)


def _pick_device(local_rank: int) -> torch.device:
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank % torch.cuda.device_count())
        return torch.device("cuda", local_rank % torch.cuda.device_count())
    return torch.device("cpu")


def init(backend: Optional[str] = None) -> Communicator:
    """Initialize from torchrun-style env (RANK/WORLD_SIZE/LOCAL_RANK)."""
    global _DEFAULT
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    device = _pick_device(local_rank)
    if world > 1 and not dist.is_initialized():
        if backend is None:
            backend = config.get("comm.backend")
        if backend == "auto":
            backend = "nccl" if device.type == "cuda" else "gloo"
        if backend == "gloo" and device.type == "cuda":
            import warnings
            warnings.warn(
                "comm backend 'gloo' with a CUDA device: gloo collectives "
                "do not support device tensors — use backend 'nccl' (RCCL) "
                "on GPUs, or hide the GPUs for CPU-only tests.",
                RuntimeWarning, stacklevel=2)
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29531")
        timeout = datetime.timedelta(
            seconds=float(config.get("comm.timeout")))
        restart = os.environ.get("TORCHELASTIC_RESTART_COUNT")
        if restart is not None:
            # under torchrun --max-restarts, namespace the store per attempt:
            # a respawned rank must never read the previous attempt's gloo
            # listener addresses (a dying predecessor's store can serve
            # stale keys for a short window -> connectFullMesh 'connection
            # refused' flakes on restart)
            store = _elastic_store(rank, world, int(restart), timeout)
            dist.init_process_group(backend=backend, rank=rank,
                                    world_size=world, store=store,
                                    timeout=timeout)
        else:
            dist.init_process_group(backend=backend, rank=rank,
                                    world_size=world, timeout=timeout)
    _DEFAULT = Communicator(rank, world, device)
    return _DEFAULT


def default() -> Communicator:
    """The process-global communicator (lazily initialized from env)."""
    global _DEFAULT
    if _DEFAULT is None:
        if int(os.environ.get("WORLD_SIZE", "1")) > 1 or dist.is_initialized():
            if dist.is_initialized():
                rank, world = dist.get_rank(), dist.get_world_size()
                _DEFAULT = Communicator(rank, world, _pick_device(
                    int(os.environ.get("LOCAL_RANK", str(rank)))))
            else:
                init()
        else:
            _DEFAULT = Communicator(0, 1, _pick_device(0))
    return _DEFAULT


def local() -> Communicator:
    """A single-process communicator (ignores any process group)."""
    return Communicator(0, 1, _pick_device(0))


def shutdown() -> None:
    global _DEFAULT
    if dist.is_initialized():
        dist.destroy_process_group()
    _DEFAULT = None
