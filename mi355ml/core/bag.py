"""DistList — a partitioned list of Python objects (text shards, sparse
blocks), the minimal dask.bag analog the text feature extractors need
(SURVEY §2.3 Bag / C12)."""

from __future__ import annotations

from typing import Callable, List, Optional, Sequence

from ..parallel import comm as comm_mod
from ..parallel.comm import Communicator


class DistList:
    """Each rank holds a local Python list; ops are SPMD."""

    def __init__(self, local: List, comm: Optional[Communicator] = None):
        self.local = list(local)
        self.comm = comm if comm is not None else comm_mod.default()

    @classmethod
    def from_global(cls, items: Sequence, comm: Optional[Communicator] = None
                    ) -> "DistList":
        comm = comm if comm is not None else comm_mod.default()
        lo, hi = comm.shard_bounds(len(items))
        return cls(list(items[lo:hi]), comm)

    def __len__(self) -> int:
        n = len(self.local)
        if self.comm.world_size == 1:
            return n
        return sum(self.comm.all_gather_object(n))

    def map(self, fn: Callable) -> "DistList":
        return DistList([fn(x) for x in self.local], self.comm)

    def map_partition(self, fn: Callable) -> "DistList":
        return DistList(fn(self.local), self.comm)

    def gather(self) -> List:
        if self.comm.world_size == 1:
            return list(self.local)
        parts = self.comm.all_gather_object(self.local)
        return [x for p in parts for x in p]

    def fold_union(self, fn: Callable) -> set:
        """Per-partition set extraction + global union (the reference's
        foldby vocabulary pattern)."""
        s = fn(self.local)
        if self.comm.world_size == 1:
            return s
        parts = self.comm.all_gather_object(sorted(s))
        out = set()
        for p in parts:
            out.update(p)
        return out
