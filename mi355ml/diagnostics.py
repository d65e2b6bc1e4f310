"""Observability (SURVEY §5.1/§5.5 rebuild): per-rank structured logging,
a lightweight op profiler (HIP events on GPU), counters, and device-memory
snapshots (amd-smi analog via torch.cuda)."""

from __future__ import annotations

import contextlib
import json
import logging
import sys
import time
from collections import defaultdict
from typing import Dict, List, Optional

import torch

from . import config
from .parallel import comm as comm_mod


def get_logger(name: str = "mi355ml") -> logging.Logger:
    log = logging.getLogger(name)
    if not log.handlers:
        h = logging.StreamHandler(sys.stderr)
        rank = comm_mod.default().rank
        h.setFormatter(logging.Formatter(
            f"[%(asctime)s rank{rank} %(levelname)s] %(message)s"))
        log.addHandler(h)
        log.setLevel(config.get("logging.level", "WARNING"))
    return log


class Profiler:
    """Context-managed per-op timeline (the reference's diagnostics.Profiler
    analog): wall times on CPU, HIP-event times on GPU."""

    def __init__(self, use_events: Optional[bool] = None):
        self.records: List[Dict] = []
        self._use_events = (torch.cuda.is_available()
                            if use_events is None else use_events)

    @contextlib.contextmanager
    def track(self, name: str, **meta):
        if self._use_events:
            s = torch.cuda.Event(enable_timing=True)
            e = torch.cuda.Event(enable_timing=True)
            s.record()
            t0 = time.perf_counter()
            try:
                yield
            finally:
                e.record()
                torch.cuda.synchronize()
                self.records.append({
                    "name": name, "gpu_ms": s.elapsed_time(e),
                    "wall_ms": (time.perf_counter() - t0) * 1e3, **meta})
        else:
            t0 = time.perf_counter()
            try:
                yield
            finally:
                self.records.append({
                    "name": name,
                    "wall_ms": (time.perf_counter() - t0) * 1e3, **meta})

    def summary(self) -> Dict[str, Dict]:
        agg: Dict[str, Dict] = defaultdict(lambda: {"n": 0, "ms": 0.0})
        for r in self.records:
            a = agg[r["name"]]
            a["n"] += 1
            a["ms"] += r.get("gpu_ms", r["wall_ms"])
        return dict(agg)

    def report(self) -> str:
        return json.dumps(self.summary(), indent=2, sort_keys=True)


class Counters:
    """Prometheus-style counters (samples/sec, collective bytes)."""

    def __init__(self):
        self.values: Dict[str, float] = defaultdict(float)

    def inc(self, name: str, v: float = 1.0):
        self.values[name] += v

    def snapshot(self) -> Dict[str, float]:
        return dict(self.values)


COUNTERS = Counters()


def device_memory() -> Dict[str, float]:
    """HBM occupancy snapshot for this rank's device (amd-smi analog)."""
    if not torch.cuda.is_available():
        return {}
    free, total = torch.cuda.mem_get_info()
    return {
        "free_gb": free / 2**30,
        "total_gb": total / 2**30,
        "allocated_gb": torch.cuda.memory_allocated() / 2**30,
        "reserved_gb": torch.cuda.memory_reserved() / 2**30,
    }


class ProgressBar:
    """Minimal terminal progress for iterative solvers (rank 0 only)."""

    def __init__(self, total: int, desc: str = "", every: int = 1):
        self.total = total
        self.desc = desc
        self.every = every
        self._rank = comm_mod.default().rank
        self._t0 = time.perf_counter()

    def update(self, i: int, **info):
        if self._rank != 0 or (i % self.every and i != self.total):
            return
        el = time.perf_counter() - self._t0
        extra = " ".join(f"{k}={v:.4g}" for k, v in info.items())
        print(f"\r{self.desc} {i}/{self.total} [{el:.1f}s] {extra}",
              end="" if i < self.total else "\n", file=sys.stderr)
