"""SparseDistArray — row-sharded CSR array (encoder outputs, sparse GLMs).

The reference's encoders emit dask arrays of scipy.sparse CSR blocks
(dask_ml/preprocessing/_encoders.py `sparse_output=True`); the MI355X
analog keeps ONE CSR shard per rank as three device tensors
(crow [m+1] / col [nnz] / vals [nnz]) and implements exactly the ops the
downstream consumers need — `dot` (X·w) and `xt_dot` (Xᵀ·r) for the GLM
solvers, `todense`/`gather`/`to_scipy` for interop — without ever
materializing the dense one-hot matrix in HBM.

SpMV runs as gather-multiply + segment-sum over the CSR structure
(`torch.segment_reduce` on the row pointer), which is deterministic
(ordered per-row reduction, no atomics) on both CPU and GPU — keeping the
framework's bitwise run-to-run guarantee for sparse fits too.
"""

from __future__ import annotations

from typing import Optional

import torch

from ..parallel import comm as comm_mod
from .array import DistArray


class SparseDistArray(DistArray):
    """Row-sharded 2-D CSR array; one (crow, col, vals) triple per rank."""

    def __init__(self, crow: torch.Tensor, col: torch.Tensor,
                 vals: torch.Tensor, n_cols: int, global_rows: int,
                 row_offset: int, comm=None):
        # deliberately NOT calling DistArray.__init__: local dense storage
        # is replaced by the CSR triple
        assert crow.dim() == col.dim() == vals.dim() == 1
        assert col.shape == vals.shape
        self.crow = crow.to(torch.int64)
        self.col = col.to(torch.int64)
        self.vals = vals
        self._ncols = int(n_cols)
        self.global_rows = int(global_rows)
        self.row_offset = int(row_offset)
        self.comm = comm if comm is not None else comm_mod.default()

    # ------------------------------------------------------------- basics
    @property
    def local(self):  # pragma: no cover - guard against dense-path misuse
        raise TypeError("SparseDistArray has no dense .local; use "
                        ".todense() or the sparse ops (dot/xt_dot)")

    @property
    def local_rows(self) -> int:
        return self.crow.shape[0] - 1

    @property
    def nnz(self) -> int:
        return self.vals.shape[0]

    @property
    def shape(self):
        return (self.global_rows, self._ncols)

    @property
    def ndim(self) -> int:
        return 2

    @property
    def ncols(self) -> int:
        return self._ncols

    @property
    def dtype(self) -> torch.dtype:
        return self.vals.dtype

    @property
    def device(self) -> torch.device:
        return self.vals.device

    def __repr__(self) -> str:  # pragma: no cover
        return (f"SparseDistArray(shape={self.shape}, nnz_local={self.nnz}, "
                f"dtype={self.dtype}, device={self.device}, "
                f"rank={self.comm.rank}/{self.comm.world_size})")

    def astype(self, dtype: torch.dtype) -> "SparseDistArray":
        return SparseDistArray(self.crow, self.col, self.vals.to(dtype),
                               self._ncols, self.global_rows,
                               self.row_offset, self.comm)

    def copy(self) -> "SparseDistArray":
        return SparseDistArray(self.crow.clone(), self.col.clone(),
                               self.vals.clone(), self._ncols,
                               self.global_rows, self.row_offset, self.comm)

    # ------------------------------------------------------------ compute
    def _row_ids(self) -> torch.Tensor:
        """[nnz] row index of every stored value (expanded crow; cached —
        the CSR structure is immutable)."""
        if getattr(self, "_row_ids_c", None) is None:
            m = self.local_rows
            lengths = self.crow[1:] - self.crow[:-1]
            self._row_ids_c = torch.repeat_interleave(
                torch.arange(m, device=self.crow.device), lengths)
        return self._row_ids_c

    def _col_groups(self):
        """Cached column-ordered (CSC-like) view for Xᵀ reductions:
        (vals_sorted, rows_sorted, uniq_cols, offsets)."""
        if getattr(self, "_col_groups_c", None) is None:
            order = torch.argsort(self.col, stable=True)
            colz = self.col[order]
            uniq, counts = torch.unique_consecutive(colz,
                                                    return_counts=True)
            offsets = torch.zeros(uniq.shape[0] + 1, dtype=torch.int64,
                                  device=self.device)
            offsets[1:] = counts.cumsum(0)
            self._col_groups_c = (self.vals[order], self._row_ids()[order],
                                  uniq, offsets)
        return self._col_groups_c

    def dot(self, w: torch.Tensor) -> DistArray:
        """X·w (replicated w) — gather-multiply + ordered per-row sum.
        Accumulates in the promoted dtype (f64 w stays f64)."""
        dt = torch.promote_types(self.vals.dtype, w.dtype)
        w = w.to(device=self.device, dtype=dt)
        vals = self.vals.to(dt)
        if w.dim() == 1:
            prod = vals * w[self.col]
        else:  # [d, k] multi-vector
            prod = vals[:, None] * w[self.col]
        z = torch.segment_reduce(prod, "sum", offsets=self.crow,
                                 axis=0, initial=0)
        return DistArray(z, self.global_rows, self.row_offset, self.comm)

    def xt_dot(self, v, dtype: torch.dtype = torch.float64) -> torch.Tensor:
        """Xᵀ·v — per-shard scatter-sum + one allreduce, replicated [d]."""
        g = self._xt_dot_local(v, dtype)
        if self.comm.world_size > 1:
            self.comm.all_reduce(g)
        return g

    def _xt_dot_local(self, v, dtype: torch.dtype = torch.float64
                      ) -> torch.Tensor:
        """Per-shard Xᵀ·v partial (no collective — solvers fuse their own)."""
        r = v.local if isinstance(v, DistArray) else v
        r = r.to(device=self.device, dtype=dtype)
        # column-sorted ordered reduction (deterministic on GPU too); the
        # CSC-like permuted view of the static structure is cached
        g = torch.zeros(self._ncols, dtype=dtype, device=self.device)
        if self.col.numel():
            vals_s, rows_s, uniq, offsets = self._col_groups()
            contrib = vals_s.to(dtype) * r[rows_s]
            sums = torch.segment_reduce(contrib, "sum",
                                        offsets=offsets, axis=0, initial=0)
            g[uniq] = sums
        return g

    def sum(self, axis: Optional[int] = 0) -> torch.Tensor:
        """Per-column value sums (category frequencies), replicated [d]."""
        if axis not in (0, None):
            raise NotImplementedError("SparseDistArray.sum supports axis=0")
        ones = torch.ones(self.local_rows, dtype=torch.float64,
                          device=self.device)
        return self.xt_dot(DistArray(ones, self.global_rows,
                                     self.row_offset, self.comm))

    # ------------------------------------------------------------ interop
    def todense(self) -> DistArray:
        m = self.local_rows
        out = torch.zeros(m, self._ncols, dtype=self.vals.dtype,
                          device=self.device)
        if self.nnz:
            out[self._row_ids(), self.col] = self.vals
        return DistArray(out, self.global_rows, self.row_offset, self.comm)

    def to_scipy(self):
        """This rank's shard as a scipy.sparse.csr_matrix (host)."""
        import scipy.sparse as sp
        return sp.csr_matrix(
            (self.vals.cpu().numpy(), self.col.cpu().numpy(),
             self.crow.cpu().numpy()),
            shape=(self.local_rows, self._ncols))

    def gather(self):
        """Full matrix on every rank as scipy CSR (tests / small results)."""
        import scipy.sparse as sp
        if self.comm.world_size == 1:
            return self.to_scipy()
        parts = self.comm.all_gather_object(self.to_scipy())
        return sp.vstack(parts).tocsr()

    def compute(self):
        return self.gather()


def from_scipy(mat, comm=None) -> SparseDistArray:
    """Shard a scipy sparse matrix by rows across ranks."""
    comm = comm if comm is not None else comm_mod.default()
    csr = mat.tocsr()
    lo, hi = comm.shard_bounds(csr.shape[0])
    part = csr[lo:hi]
    dev = comm.device
    return SparseDistArray(
        torch.from_numpy(part.indptr.astype("int64")).to(dev),
        torch.from_numpy(part.indices.astype("int64")).to(dev),
        torch.from_numpy(part.data).to(dev),
        csr.shape[1], csr.shape[0], lo, comm)


def _exchange_blobs(comm, blobs):
    """True point-to-point exchange of per-destination pickled blobs
    (the C8 gather_dep analog; same pattern as the dataframe shuffle):
    one isend/irecv per nonempty (src, dst) pair over the direct link.
    Returns {src_rank: bytes} for this rank (own blob included)."""
    import torch.distributed as dist
    W = comm.world_size
    sizes_local = torch.tensor([len(b) for b in blobs], dtype=torch.int64)
    sizes = [s.cpu() for s in comm.all_gather(
        sizes_local.to(comm.device) if comm.device.type == "cuda"
        else sizes_local)]
    ops = []
    recv_bufs = {}
    for r in range(W):
        if r == comm.rank:
            continue
        if len(blobs[r]):
            ops.append(dist.P2POp(dist.isend, torch.frombuffer(
                bytearray(blobs[r]), dtype=torch.uint8), r))
        n_in = int(sizes[r][comm.rank])
        if n_in:
            buf = torch.empty(n_in, dtype=torch.uint8)
            recv_bufs[r] = buf
            ops.append(dist.P2POp(dist.irecv, buf, r))
    if ops:
        for req in dist.batch_isend_irecv(ops):
            req.wait()
    out = {}
    if blobs[comm.rank]:
        out[comm.rank] = bytes(blobs[comm.rank])
    for r, buf in recv_bufs.items():
        out[r] = buf.numpy().tobytes()
    return out


def _csr_slice(A: SparseDistArray, lo: int, hi: int):
    """Host-side (crow, col, vals) numpy triple for local rows [lo, hi)."""
    p0 = int(A.crow[lo])
    p1 = int(A.crow[hi])
    return ((A.crow[lo:hi + 1] - p0).cpu().numpy(),
            A.col[p0:p1].cpu().numpy(),
            A.vals[p0:p1].cpu().numpy())


def rebalance_sparse(A: SparseDistArray) -> SparseDistArray:
    """Redistribute CSR rows to the canonical balanced contiguous layout
    (the dense DistArray.rebalance analog — VERDICT r1 #10 sparse interop).
    Rows move as pickled CSR slices over point-to-point sends; nnz per row
    is ragged, so this is the C4-style ragged payload path, not the dense
    tensor p2p."""
    import pickle
    comm = A.comm
    W = comm.world_size
    if W == 1:
        return SparseDistArray(A.crow, A.col, A.vals, A._ncols,
                               A.global_rows, 0, comm)
    dev = A.device
    n_local = torch.tensor([A.local_rows], dtype=torch.int64,
                           device=dev if dev.type == "cuda" else "cpu")
    counts = [int(c.item()) for c in comm.all_gather(n_local)]
    total = sum(counts)
    cur_start = sum(counts[:comm.rank])
    base, rem = divmod(total, W)
    tgt_starts = []
    acc = 0
    for r in range(W):
        tgt_starts.append(acc)
        acc += base + (1 if r < rem else 0)
    tgt_starts.append(acc)
    cs, ce = cur_start, cur_start + A.local_rows
    blobs = []
    for r in range(W):
        lo = max(cs, tgt_starts[r])
        hi = min(ce, tgt_starts[r + 1])
        blobs.append(pickle.dumps(_csr_slice(A, lo - cs, hi - cs))
                     if hi > lo else b"")
    parts = _exchange_blobs(comm, blobs)
    # assemble in src-rank order (ranks hold ascending global row ranges)
    crows, cols, vals = [], [], []
    nnz_off = 0
    row_cnt = 0
    for r in sorted(parts):
        crow, col, v = pickle.loads(parts[r])
        crows.append(torch.from_numpy(crow[1:].copy()) + nnz_off)
        cols.append(torch.from_numpy(col.copy()))
        vals.append(torch.from_numpy(v.copy()))
        nnz_off += len(col)
        row_cnt += len(crow) - 1
    my_rows = tgt_starts[comm.rank + 1] - tgt_starts[comm.rank]
    assert row_cnt == my_rows, (row_cnt, my_rows)
    crow = torch.zeros(my_rows + 1, dtype=torch.int64)
    if crows:
        crow[1:] = torch.cat(crows)
    col = (torch.cat(cols) if cols
           else torch.empty(0, dtype=torch.int64))
    v = torch.cat(vals) if vals else torch.empty(0, dtype=A.vals.dtype)
    return SparseDistArray(crow.to(dev), col.to(dev), v.to(dev), A._ncols,
                           A.global_rows, tgt_starts[comm.rank], comm)


# override the dense p2p rebalance (which needs .local) with the CSR form
SparseDistArray.rebalance = rebalance_sparse


class SpilledSparseDistArray(SparseDistArray):
    """CSR shard resident in (pinned) host memory, streamed to the device
    in row blocks (the sparse member of the spill tier — core/spill.py's
    dense design applied to the (crow, col, vals) triple). SpMV-consuming
    code (GLM sparse value+grad, column sums) works unchanged: dot /
    _xt_dot_local / sum accumulate over the streamed blocks with the same
    deterministic segment reductions."""

    def __init__(self, crow, col, vals, n_cols, global_rows, row_offset,
                 comm=None, block_rows=None):
        super().__init__(crow, col, vals, n_cols, global_rows, row_offset,
                         comm)
        assert self.vals.device.type == "cpu", "spilled storage is host"
        from .. import config as _config
        self.block_rows = block_rows or int(_config.get("array.block_rows"))
        self._copy_stream = (torch.cuda.Stream(self.comm.device)
                             if self.comm.device.type == "cuda" else None)

    @property
    def device(self):  # compute device, not storage
        return self.comm.device

    def device_blocks(self):
        """Yield (row_lo, SparseDistArray device block); double-buffered
        H2D of the three CSR tensors on a copy stream."""
        dev = self.comm.device
        m = self.local_rows
        bs = self.block_rows
        starts = list(range(0, max(m, 1), bs)) if m else []

        def host_block(lo):
            hi = min(m, lo + bs)
            p0 = int(self.crow[lo])
            p1 = int(self.crow[hi])
            return (self.crow[lo:hi + 1] - p0, self.col[p0:p1],
                    self.vals[p0:p1], hi - lo)

        if dev.type != "cuda":
            for lo in starts:
                crow_b, col_b, vals_b, rows = host_block(lo)
                yield lo, SparseDistArray(crow_b, col_b, vals_b,
                                          self._ncols, rows, 0, self.comm)
            return
        bufs = [None, None]
        evts = [torch.cuda.Event(), torch.cuda.Event()]

        def _issue(i, slot):
            crow_b, col_b, vals_b, rows = host_block(starts[i])
            with torch.cuda.stream(self._copy_stream):
                bufs[slot] = (crow_b.to(dev, non_blocking=True),
                              col_b.to(dev, non_blocking=True),
                              vals_b.to(dev, non_blocking=True), rows)
                evts[slot].record(self._copy_stream)

        if starts:
            _issue(0, 0)
        for i, lo in enumerate(starts):
            slot = i & 1
            if i + 1 < len(starts):
                _issue(i + 1, slot ^ 1)
            torch.cuda.current_stream(dev).wait_event(evts[slot])
            crow_b, col_b, vals_b, rows = bufs[slot]
            yield lo, SparseDistArray(crow_b, col_b, vals_b, self._ncols,
                                      rows, 0, self.comm)
            evts[slot].record(torch.cuda.current_stream(dev))
            self._copy_stream.wait_event(evts[slot])

    # -- streamed SpMV ops -------------------------------------------------
    def dot(self, w: torch.Tensor) -> "DistArray":
        dt = torch.promote_types(self.vals.dtype, w.dtype)
        dev = self.comm.device
        w = w.to(device=dev, dtype=dt)
        out = None
        for lo, blk in self.device_blocks():
            z = blk.dot(w).local
            if out is None:
                shape = (self.local_rows,) + tuple(z.shape[1:])
                out = torch.zeros(shape, dtype=z.dtype, device=dev)
            out[lo:lo + z.shape[0]] = z
        if out is None:
            out = torch.zeros(0, dtype=dt, device=dev)
        from .array import DistArray
        return DistArray(out, self.global_rows, self.row_offset, self.comm)

    def _xt_dot_local(self, v, dtype: torch.dtype = torch.float64):
        r = v.local if hasattr(v, "local") else v
        r = r.to(device=self.comm.device, dtype=dtype)
        g = torch.zeros(self._ncols, dtype=dtype, device=self.comm.device)
        for lo, blk in self.device_blocks():
            g += blk._xt_dot_local(r[lo:lo + blk.local_rows], dtype)
        return g

    def sum(self, axis=0):
        if axis not in (0, None):
            raise NotImplementedError("sum supports axis=0")
        g = torch.zeros(self._ncols, dtype=torch.float64,
                        device=self.comm.device)
        for _, blk in self.device_blocks():
            ones = torch.ones(blk.local_rows, dtype=torch.float64,
                              device=self.comm.device)
            g += blk._xt_dot_local(ones)  # local partial; ONE reduce below
        if self.comm.world_size > 1:
            self.comm.all_reduce(g)
        return g


def spill_sparse(S: SparseDistArray,
                 block_rows=None) -> SpilledSparseDistArray:
    """Move a device CSR shard into (pinned) host memory for streaming."""
    def host(t):
        t = t.cpu()
        if torch.cuda.is_available():
            t = t.pin_memory()
        return t
    return SpilledSparseDistArray(host(S.crow), host(S.col), host(S.vals),
                                  S._ncols, S.global_rows, S.row_offset,
                                  S.comm, block_rows)
