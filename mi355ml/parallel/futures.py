"""SPMD futures pool — the Client.submit/gather analog (SURVEY §2.8
"Task parallel": delayed/futures, distributed/client.py:1473).

The reference's futures live on a scheduler that pushes tasks to workers.
In an SPMD runtime there is no scheduler process, so the pool works by
CONVENTION: every rank calls submit() in the same order (deterministic task
ids + round-robin owners), and run() is a collective step in which each
rank executes the tasks it owns and publishes the pickled results through
the control-plane store; Future.result() then reads the value on every
rank. Exceptions are captured on the owner and re-raised everywhere.

    pool = FuturePool()
    futs = [pool.submit(fn, arg) for arg in args]   # all ranks, same order
    pool.run()                                      # collective
    values = [f.result() for f in futs]
"""

from __future__ import annotations

import pickle
from datetime import timedelta
from typing import Any, Callable, List, Optional

from . import comm as comm_mod
from .coordination import get_store


class _RemoteError(Exception):
    pass


class Future:
    def __init__(self, pool: "FuturePool", tid: int, owner: int):
        self._pool = pool
        self.tid = tid
        self.owner = owner

    def done(self) -> bool:
        return self.tid in self._pool._results or \
            get_store().check([self._pool._key(self.tid)])

    def result(self, timeout: float = 300.0) -> Any:
        if self.tid in self._pool._pending:
            self._pool.run()
        val = self._pool._fetch(self.tid, timeout)
        if isinstance(val, _RemoteError):
            raise RuntimeError(
                f"task {self.tid} failed on rank {self.owner}: "
                f"{val.args[0]}")
        return val


class FuturePool:
    def __init__(self, name: str = "pool", comm=None):
        self.comm = comm or comm_mod.default()
        self.name = name
        self._next = 0
        self._pending = {}  # tid -> (fn, args, kwargs, owner)
        self._results = {}  # local cache

    def _key(self, tid: int) -> str:
        return f"mi355ml/fut/{self.name}/{tid}"

    def submit(self, fn: Callable, *args, rank: Optional[int] = None,
               **kwargs) -> Future:
        tid = self._next
        self._next += 1
        owner = rank if rank is not None else tid % self.comm.world_size
        self._pending[tid] = (fn, args, kwargs, owner)
        return Future(self, tid, owner)

    def map(self, fn: Callable, iterable) -> List[Future]:
        return [self.submit(fn, x) for x in iterable]

    def run(self) -> None:
        """Collective: execute pending tasks (owners only), publish results."""
        store = get_store()
        pending, self._pending = self._pending, {}
        for tid, (fn, args, kwargs, owner) in pending.items():
            if owner == self.comm.rank:
                try:
                    val = fn(*args, **kwargs)
                except Exception as e:  # ship the failure, not a hang
                    val = _RemoteError(repr(e))
                self._results[tid] = val
                if self.comm.world_size > 1:
                    store.set(self._key(tid), pickle.dumps(val))
            elif self.comm.world_size == 1:
                raise RuntimeError("task owned by absent rank")

    def _fetch(self, tid: int, timeout: float) -> Any:
        if tid in self._results:
            return self._results[tid]
        store = get_store()
        store.wait([self._key(tid)], timedelta(seconds=timeout))
        val = pickle.loads(bytes(store.get(self._key(tid))))
        self._results[tid] = val
        return val

    def gather(self, futures: List[Future], timeout: float = 300.0) -> List:
        self.run()
        return [f.result(timeout) for f in futures]
