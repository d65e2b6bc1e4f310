"""Column-sharded distributed matrix for very wide X (SURVEY §5.7 /
§2.8 "model/feature parallel"; reference analog: 2-D block algebra in
dask/array/routines.py:275 tensordot and linalg.py:508 sfqr).

Layout: every rank holds ALL m rows and a contiguous slice of the feature
axis — the right shard for X wider than ~50k columns, where a row-shard
would make the per-iteration `w` broadcast and `Xᵀr` reduce dominate.

Communication pattern (the feature-parallel duals of the row-shard ops):
  X @ w      per-rank  [m, d_loc] @ [d_loc]  partial  →  allreduce [m]
  Xᵀ r       per-rank  [d_loc] slice, NO communication (already sharded)
  full grad  allgather_varlen of the [d_loc] slices → [d]
  SVD        randomized range finder with col-sharded Ω: Y = Σ X_loc Ω_loc
             (allreduce [m, l]), B = Qᵀ X col-sharded, small Gram eigh.
"""

from __future__ import annotations

from typing import Optional, Tuple

import numpy as np
import torch

from ..parallel.comm import Communicator
from ..parallel import comm as comm_mod


class ColDistArray:
    """[m, d] matrix column-sharded across ranks; local = [m, d_local]."""

    def __init__(self, local: torch.Tensor, global_cols: int,
                 col_offset: int, comm: Communicator):
        assert local.dim() == 2
        self.local = local
        self.global_cols = int(global_cols)
        self.col_offset = int(col_offset)
        self.comm = comm

    # -- descriptors ------------------------------------------------------
    @property
    def shape(self) -> Tuple[int, int]:
        return (self.local.shape[0], self.global_cols)

    @property
    def ndim(self) -> int:
        return 2

    @property
    def ncols(self) -> int:
        return self.global_cols

    @property
    def local_cols(self) -> int:
        return self.local.shape[1]

    @property
    def global_rows(self) -> int:
        return self.local.shape[0]

    @property
    def dtype(self) -> torch.dtype:
        return self.local.dtype

    @property
    def device(self) -> torch.device:
        return self.local.device

    def __len__(self) -> int:
        return self.local.shape[0]

    def __repr__(self) -> str:  # pragma: no cover
        return (f"ColDistArray(shape={self.shape}, "
                f"cols[{self.col_offset}:"
                f"{self.col_offset + self.local_cols}], "
                f"rank={self.comm.rank}/{self.comm.world_size})")

    def astype(self, dtype: torch.dtype) -> "ColDistArray":
        return ColDistArray(self.local.to(dtype), self.global_cols,
                            self.col_offset, self.comm)

    # -- feature-parallel linear algebra ----------------------------------
    def dot(self, w: torch.Tensor) -> torch.Tensor:
        """X @ w for a FULL-length (replicated) w: local partial GEMV +
        one allreduce of the [m] partials. Returns the replicated [m]
        (or [m, p]) product on every rank."""
        wl = self._slice_cols(w)
        z = self.local @ wl
        if self.comm.world_size > 1:
            self.comm.all_reduce(z)
        return z

    def rmatvec_local(self, r: torch.Tensor) -> torch.Tensor:
        """Xᵀ r: the [d_local] slice this rank owns — no communication."""
        return self.local.t() @ r

    def rmatvec(self, r: torch.Tensor) -> torch.Tensor:
        """Full [d] (or [d, p]) Xᵀ r via local slice + allgather."""
        return self.gather_cols(self.rmatvec_local(r))

    def gram_small(self, B_local: torch.Tensor,
                   dtype: torch.dtype = torch.float64) -> torch.Tensor:
        """Σ_ranks B_loc B_locᵀ for a col-sharded [l, d_local] panel —
        one [l, l] allreduce (the wide-X TSQR/CAQR analog's small Gram)."""
        g = B_local.to(dtype) @ B_local.to(dtype).t()
        if self.comm.world_size > 1:
            self.comm.all_reduce(g)
        return g

    def gather_cols(self, part: torch.Tensor) -> torch.Tensor:
        """Concatenate per-rank column-slice results (first dim = cols)."""
        if self.comm.world_size == 1:
            return part
        parts = self.comm.all_gather_varlen(part.contiguous())
        return torch.cat(parts, dim=0)

    def _slice_cols(self, w: torch.Tensor) -> torch.Tensor:
        if w.shape[0] == self.local_cols and self.global_cols != \
                self.local_cols:
            return w  # already the local slice
        return w[self.col_offset:self.col_offset + self.local_cols]

    # -- per-column stats (no communication: columns are local) -----------
    def mean(self, axis: int = 0) -> torch.Tensor:
        assert axis == 0
        return self.gather_cols(self.local.mean(dim=0))

    def var(self, axis: int = 0, ddof: int = 0) -> torch.Tensor:
        assert axis == 0
        m = self.local.shape[0]
        v = self.local.var(dim=0, unbiased=False)
        if ddof:
            v = v * (m / max(m - ddof, 1))
        return self.gather_cols(v)

    def min(self, axis: int = 0) -> torch.Tensor:
        assert axis == 0
        return self.gather_cols(self.local.min(dim=0).values)

    def max(self, axis: int = 0) -> torch.Tensor:
        assert axis == 0
        return self.gather_cols(self.local.max(dim=0).values)

    # -- materialization ---------------------------------------------------
    def gather(self) -> torch.Tensor:
        """Full [m, d] on every rank (tests/oracles only)."""
        if self.comm.world_size == 1:
            return self.local
        parts = self.comm.all_gather_varlen(
            self.local.t().contiguous())
        return torch.cat(parts, dim=0).t().contiguous()

    def to_numpy(self) -> np.ndarray:
        return self.gather().cpu().numpy()


class Replicated:
    """Per-row result of a feature-parallel op: the SAME full-length [m]
    (or [m, p]) tensor on every rank (e.g. sigmoid(X·w) for col-sharded X).
    check_array() converts it to a row-sharded DistArray when it meets
    row-sharded code (metrics, downstream estimators)."""

    def __init__(self, t: torch.Tensor, comm: Communicator):
        self.local = t
        self.comm = comm

    @property
    def shape(self):
        return tuple(self.local.shape)

    @property
    def ncols(self) -> int:
        return self.local.shape[1] if self.local.dim() > 1 else 1

    @property
    def dtype(self):
        return self.local.dtype

    @property
    def device(self):
        return self.local.device

    def map_local(self, fn) -> "Replicated":
        return Replicated(fn(self.local), self.comm)

    def gather(self) -> torch.Tensor:
        return self.local

    def to_numpy(self) -> np.ndarray:
        return self.local.cpu().numpy()


def col_shard_bounds(ncols: int, comm: Communicator) -> Tuple[int, int]:
    base = ncols // comm.world_size
    extra = ncols % comm.world_size
    lo = comm.rank * base + min(comm.rank, extra)
    hi = lo + base + (1 if comm.rank < extra else 0)
    return lo, hi


def shard_cols(full: torch.Tensor,
               comm: Optional[Communicator] = None) -> ColDistArray:
    """Shard a replicated [m, d] tensor by columns (every rank passes the
    same full tensor, keeps its slice)."""
    comm = comm or comm_mod.default()
    lo, hi = col_shard_bounds(full.shape[1], comm)
    local = full[:, lo:hi].contiguous().to(comm.device)
    return ColDistArray(local, full.shape[1], lo, comm)


def from_numpy_cols(a: np.ndarray,
                    comm: Optional[Communicator] = None) -> ColDistArray:
    return shard_cols(torch.from_numpy(np.ascontiguousarray(a)), comm)
