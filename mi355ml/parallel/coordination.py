"""Distributed coordination primitives (SURVEY §2.5 "Coordination":
distributed Event/Lock/Variable/Queue/Semaphore scheduler extensions).

The reference implements these as scheduler-extension KV state over the
RPC plane. Here they ride the control plane we already have: the process
group's TCPStore (rank 0 hosts it) — set/get/add/compare_set/wait are
atomic server-side ops, which is exactly the reference's semantics. With
world_size == 1 an in-process dict serves the same API.

These are CONTROL-plane primitives (trial orchestration, debugging) — the
data plane stays RCCL collectives.
"""

from __future__ import annotations

import pickle
import threading
import time
from datetime import timedelta
from typing import Any, Optional

import torch.distributed as dist

from . import comm as comm_mod


class _LocalStore:
    """world_size==1 stand-in with TCPStore-compatible ops."""

    def __init__(self):
        self._d = {}
        self._lock = threading.Lock()

    def set(self, k, v):
        with self._lock:
            self._d[k] = v if isinstance(v, bytes) else str(v).encode()

    def get(self, k):
        deadline = time.monotonic() + 30
        while True:
            with self._lock:
                if k in self._d:
                    return self._d[k]
            if time.monotonic() > deadline:
                raise KeyError(k)
            time.sleep(0.01)

    def add(self, k, v):
        with self._lock:
            cur = int(self._d.get(k, b"0"))
            cur += v
            self._d[k] = str(cur).encode()
            return cur

    def compare_set(self, k, expected, desired):
        with self._lock:
            cur = self._d.get(k, b"")
            exp = expected if isinstance(expected, bytes) else str(expected).encode()
            des = desired if isinstance(desired, bytes) else str(desired).encode()
            if (k not in self._d and exp == b"") or cur == exp:
                self._d[k] = des
                return des
            return cur

    def wait(self, keys, timeout=None):
        deadline = time.monotonic() + (timeout.total_seconds()
                                       if timeout else 300)
        while True:
            with self._lock:
                if all(k in self._d for k in keys):
                    return
            if time.monotonic() > deadline:
                raise TimeoutError(keys)
            time.sleep(0.01)

    def check(self, keys):
        with self._lock:
            return all(k in self._d for k in keys)


_local_store = _LocalStore()


def get_store():
    if dist.is_initialized():
        return dist.distributed_c10d._get_default_store()
    return _local_store


class Variable:
    """Shared named value (distributed/variable.py:126 parity)."""

    def __init__(self, name: str = "variable"):
        self._key = f"mi355ml/var/{name}"

    def set(self, value: Any) -> None:
        get_store().set(self._key, pickle.dumps(value))

    def get(self, timeout: Optional[float] = None) -> Any:
        store = get_store()
        if not store.check([self._key]):
            # never fall into the store's long blocking-get default
            if timeout is None:
                raise KeyError(self._key)
            store.wait([self._key], timedelta(seconds=timeout))
        return pickle.loads(bytes(store.get(self._key)))


class Event:
    """Distributed event flag (distributed/event.py:153 parity)."""

    def __init__(self, name: str = "event"):
        self._key = f"mi355ml/event/{name}"

    def set(self) -> None:
        get_store().set(self._key, b"1")

    def is_set(self) -> bool:
        try:
            s = get_store()
            if isinstance(s, _LocalStore):
                return self._key in s._d
            return bytes(s.compare_set(self._key, "", "")) not in (b"", b"0")
        except Exception:
            return False

    def wait(self, timeout: float = 300.0) -> None:
        get_store().wait([self._key], timedelta(seconds=timeout))

    def clear(self) -> None:
        get_store().set(self._key, b"0")


class Lock:
    """Distributed mutex via atomic compare_set (distributed/lock.py:76)."""

    def __init__(self, name: str = "lock"):
        self._key = f"mi355ml/lock/{name}"
        self._id = f"{comm_mod.default().rank}:{threading.get_ident()}"

    def acquire(self, timeout: float = 300.0) -> bool:
        store = get_store()
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            got = bytes(store.compare_set(self._key, "", self._id))
            if got in (b"", self._id.encode()):
                return True
            time.sleep(0.005)
        return False

    def release(self) -> None:
        store = get_store()
        store.compare_set(self._key, self._id, "")

    def __enter__(self):
        if not self.acquire():
            raise TimeoutError("lock acquire timed out")
        return self

    def __exit__(self, *exc):
        self.release()


class Semaphore:
    """Counting semaphore (distributed/semaphore.py:272 parity)."""

    def __init__(self, max_leases: int = 1, name: str = "semaphore"):
        self._key = f"mi355ml/sem/{name}"
        self.max_leases = max_leases

    def acquire(self, timeout: float = 300.0) -> bool:
        store = get_store()
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            cur = store.add(self._key, 1)
            if cur <= self.max_leases:
                return True
            store.add(self._key, -1)
            time.sleep(0.005)
        return False

    def release(self) -> None:
        get_store().add(self._key, -1)

    def __enter__(self):
        if not self.acquire():
            raise TimeoutError("semaphore acquire timed out")
        return self

    def __exit__(self, *exc):
        self.release()


class Queue:
    """Multi-producer multi-consumer FIFO (distributed/queues.py:130).

    Store-backed: a tail counter allocates slots (atomic add), a head
    counter allocates reads; each slot is one pickled value.
    """

    def __init__(self, name: str = "queue"):
        self._base = f"mi355ml/queue/{name}"

    def put(self, value: Any) -> None:
        store = get_store()
        slot = store.add(f"{self._base}/tail", 1)
        store.set(f"{self._base}/item/{slot}", pickle.dumps(value))

    def get(self, timeout: float = 300.0) -> Any:
        store = get_store()
        slot = store.add(f"{self._base}/head", 1)
        key = f"{self._base}/item/{slot}"
        store.wait([key], timedelta(seconds=timeout))
        return pickle.loads(bytes(store.get(key)))

    def qsize(self) -> int:
        store = get_store()
        tail = store.add(f"{self._base}/tail", 0)
        head = store.add(f"{self._base}/head", 0)
        return max(0, int(tail) - int(head))
