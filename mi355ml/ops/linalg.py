"""Distributed tall-skinny linear algebra (K10–K12, C5–C6 in SURVEY §2.7).

tsqr: the reference's recursive blockwise TSQR graph
(dask/array/linalg.py:52, algorithm in SURVEY A.1) mapped to one node:
stage-1 panel QRs run per ROW BLOCK of the local shard (bounded blocks keep
rocSOLVER inside int32 index space and bound the float64 staging memory —
this is exactly the reference's recursion step A.1-2), the per-rank R
stacks reduce locally, then ONE all-gather of the W x [d, d] R panels (C5),
a redundant-deterministic QR/SVD of the stacked R on every rank (no
broadcast needed), and local Q1 @ Q2 GEMMs for Q only when the caller needs
it (PCA/TSVD fit paths don't).

svd_compressed: Halko randomized range finder
(dask/array/linalg.py:616-740): shared Ω from the master seed, per-rank f32
GEMMs (the small projected factors are refined in f64), TSQR
re-orthogonalization per power iteration, allreduce for the XᵀQ
projections (C6).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from ..core.array import DistArray
from ..core import random as mlrand


def _block_rows_for(d: int) -> int:
    """Stage-1 row-block size: keep the f64 staging block <= ~4 GB and the
    element count well under 2^31 (rocSOLVER int32 indexing)."""
    return max(d, min(4_000_000, (1 << 30) // max(d, 1)))


def _local_qr(a: torch.Tensor, d: int, keep_q: bool
              ) -> Tuple[Optional[torch.Tensor], torch.Tensor]:
    """Reduced QR of one tall block.

    Primary path: CholeskyQR(2) — the Gram matrix comes from one GEMM
    (matrix-core work instead of rocSOLVER's latency-bound Householder
    panels, ~500x faster on [1e6 x 1000] blocks), Cholesky runs on the tiny
    [d, d] f64 Gram, and when Q is needed a second pass (CholeskyQR2)
    restores orthogonality to f32 roundoff. Falls back to torch.linalg.qr
    (rocSOLVER/LAPACK) when the Gram is numerically rank-deficient (e.g.
    power-iteration panels with astronomical condition numbers).
    """
    m = a.shape[0]
    if m < d:
        pad = torch.zeros(d - m, d, dtype=a.dtype, device=a.device)
        q, r = torch.linalg.qr(torch.cat([a, pad], dim=0), mode="reduced")
        return (q[:m] if keep_q else None), r
    try:
        # Gram in the INPUT dtype (rocBLAS f64 GEMM runs at ~20 GFLOPS on
        # these thin shapes — 70x slower than sgemm); only the tiny [d, d]
        # result is upcast for the Cholesky. CholeskyQR2's second pass
        # restores Q orthogonality; a ~sqrt(n)*eps_f32 Gram perturbation
        # moves singular values ~1e-4 relative, inside this framework's
        # fp32 tolerance class.
        g = (a.t() @ a).double()
        r1 = torch.linalg.cholesky(g, upper=True)  # XᵀX = R1ᵀ R1
        if not keep_q:
            return None, r1
        rinv1 = torch.linalg.solve_triangular(
            r1, torch.eye(d, dtype=torch.float64, device=a.device),
            upper=True)
        q1 = a @ rinv1.to(a.dtype)
        # second pass: re-orthogonalize (CholeskyQR2)
        g2 = (q1.t() @ q1).double()
        r2 = torch.linalg.cholesky(g2, upper=True)
        rinv2 = torch.linalg.solve_triangular(
            r2, torch.eye(d, dtype=torch.float64, device=a.device),
            upper=True)
        q = q1 @ rinv2.to(a.dtype)
        return q, r2 @ r1
    except torch.linalg.LinAlgError:
        q, r = torch.linalg.qr(a, mode="reduced")
        return (q if keep_q else None), r


def _shard_tsqr_r(a: torch.Tensor, d: int, keep_q: bool):
    """Blocked stage-1 TSQR of the local shard.

    Returns (blocks, r_local): blocks is a list of
    (row_lo, Q1_block f32 or None, R1 [d,d] f64) and r_local the [d, d]
    float64 R of the whole shard; when the shard needed several blocks an
    inner stacked QR combines them (A.1 recursion, depth 1 suffices)."""
    m = a.shape[0]
    br = _block_rows_for(d)
    blocks = []
    rs = []
    for lo in range(0, max(m, 1), br):
        blk = a[lo:lo + br]
        q, r = _local_qr(blk, d, keep_q)
        blocks.append([lo, q, r.to(torch.float64)])
        rs.append(r.to(torch.float64))
        del q
    if len(rs) == 1:
        return blocks, rs[0], None
    stacked = torch.cat(rs, dim=0)  # [nb*d, d] f64 (small)
    q2, r = torch.linalg.qr(stacked, mode="reduced")
    return blocks, r, q2  # q2: [nb*d, d]


def tsqr(X: DistArray, compute_svd: bool = False, compute_q: bool = True):
    """QR (or SVD) of a tall-skinny row-sharded matrix.

    Returns (Q DistArray, R) / (U DistArray, s, Vh); with compute_q=False
    the Q/U factor is skipped (None) — R/s/Vh are replicated and exact.
    """
    a = X._as2d()
    d = a.shape[1]
    comm = X.comm
    blocks, r_local, q2_inner = _shard_tsqr_r(a, d, keep_q=compute_q)
    if comm.world_size > 1:
        rs = comm.all_gather(r_local.contiguous())
        stacked = torch.cat(rs, dim=0)  # [W*d, d]
    else:
        stacked = r_local
    q3, r = torch.linalg.qr(stacked, mode="reduced")  # replicated determin.
    q3_own = q3[comm.rank * d:(comm.rank + 1) * d]  # this rank's block of Q3

    def _assemble(right: torch.Tensor) -> torch.Tensor:
        """Q_local @ right for the factored Q_local (blocked backprop)."""
        out = torch.empty(a.shape[0], right.shape[1], dtype=X.dtype,
                          device=a.device)
        for i, (lo, q1, _) in enumerate(blocks):
            piece = right if q2_inner is None else \
                (q2_inner[i * d:(i + 1) * d] @ right)
            out[lo:lo + q1.shape[0]] = q1 @ piece.to(X.dtype)
        return out

    if not compute_svd:
        if not compute_q:
            return None, r.to(X.dtype)
        q = _assemble(q3_own)
        return (DistArray(q, X.global_rows, X.row_offset, comm),
                r.to(X.dtype))
    u_r, s, vh = torch.linalg.svd(r, full_matrices=False)
    # deterministic signs (svd_flip, V-based): largest |Vh| entry positive
    idx = vh.abs().argmax(dim=1)
    signs = torch.sign(vh[torch.arange(vh.shape[0], device=vh.device), idx])
    signs[signs == 0] = 1.0
    vh = vh * signs[:, None]
    u_r = u_r * signs[None, :]
    if not compute_q:
        return None, s.to(X.dtype), vh.to(X.dtype)
    u = _assemble(q3_own @ u_r)
    return (DistArray(u, X.global_rows, X.row_offset, comm),
            s.to(X.dtype), vh.to(X.dtype))


def svd_compressed(X: DistArray, k: int, n_power_iter: int = 2,
                   n_oversamples: int = 10, seed: int = 0,
                   compute_u: bool = True):
    """Randomized truncated SVD (Halko/Martinsson/Tropp), f32 GEMMs.

    Returns (U DistArray [m, k] or None, s [k], Vh [k, d]).
    """
    a = X._as2d()
    m, d = X.global_rows, a.shape[1]
    comm = X.comm
    ell = min(k + n_oversamples, d, m)
    g = mlrand.generator(seed, shard_id=0xABCD, device=a.device)
    omega = torch.randn(d, ell, generator=g, device=a.device, dtype=X.dtype)
    Y = DistArray(a @ omega, m, X.row_offset, comm)
    Q, _ = tsqr(Y)
    for _ in range(n_power_iter):
        # Z = Xᵀ Q  (allreduce, C6), then re-orthogonalize X·Z
        z = a.t() @ Q._as2d()
        if comm.world_size > 1:
            comm.all_reduce(z)
        Y = DistArray(a @ z, m, X.row_offset, comm)
        Q, _ = tsqr(Y)
    # B = Qᵀ X : [ell, d] via allreduce (f64 for the small final factor)
    b = (Q._as2d().t() @ a).to(torch.float64)
    if comm.world_size > 1:
        comm.all_reduce(b)
    u_b, s, vh = torch.linalg.svd(b, full_matrices=False)
    idx = vh.abs().argmax(dim=1)
    signs = torch.sign(vh[torch.arange(vh.shape[0], device=vh.device), idx])
    signs[signs == 0] = 1.0
    vh = vh * signs[:, None]
    u_b = u_b * signs[None, :]
    u = None
    if compute_u:
        u = DistArray(Q._as2d() @ u_b[:, :k].to(X.dtype), m, X.row_offset,
                      comm)
    return u, s[:k].to(X.dtype), vh[:k].to(X.dtype)


def svd_randomized_col(X, k: int, n_power_iter: int = 2,
                       n_oversamples: int = 10, seed: int = 0,
                       compute_u: bool = True):
    """Randomized truncated SVD for a COLUMN-sharded matrix
    (core.colarray.ColDistArray — the wide-X mechanism).

    Feature-parallel duals of svd_compressed's collectives:
      range finder   Y = Σ_r X_loc Ω_loc        → allreduce [m, ell]
      power iter     B_loc = X_locᵀ Q (local);  Y = Σ X_loc B_loc → allreduce
      final factor   G = Σ B_loc B_locᵀ         → allreduce [ell, ell] f64
      eigh(G) replicated-deterministic; Vh comes out column-sharded.

    Q lives replicated ([m, ell] on every rank, orthonormalized by a local
    QR — identical on all ranks); no TSQR needed because the row axis is
    not sharded. Returns (U [m, k] replicated tensor or None, s [k],
    Vh_full [k, d] gathered tensor).
    """
    a = X.local
    m, d = X.global_rows, X.global_cols
    comm = X.comm
    ell = min(k + n_oversamples, d, m)
    # full Ω generated from the seed, each rank keeps its row slice —
    # bit-identical results regardless of world size
    g = mlrand.generator(seed, shard_id=0xC01, device=a.device)
    omega = torch.randn(d, ell, generator=g, device=a.device, dtype=X.dtype)
    om_loc = omega[X.col_offset:X.col_offset + X.local_cols]
    Y = a @ om_loc
    if comm.world_size > 1:
        comm.all_reduce(Y)
    Q, _ = torch.linalg.qr(Y)
    for _ in range(n_power_iter):
        b_loc = a.t() @ Q          # [d_loc, ell], no comm
        Y = a @ b_loc
        if comm.world_size > 1:
            comm.all_reduce(Y)
        Q, _ = torch.linalg.qr(Y)
    b_loc = (Q.t() @ a).to(torch.float64)      # [ell, d_loc], no comm
    gram = b_loc @ b_loc.t()
    if comm.world_size > 1:
        comm.all_reduce(gram)
    evals, evecs = torch.linalg.eigh(gram)     # ascending
    evals = evals.flip(0).clamp(min=0.0)
    u_b = evecs.flip(1)                        # [ell, ell] descending
    s = evals.sqrt()
    safe = s.clamp(min=torch.finfo(torch.float64).tiny)
    vh_loc = (u_b.t() @ b_loc) / safe[:, None]  # [ell, d_loc] col-sharded
    # svd_flip parity: sign from the globally largest-|v| entry per row
    vh_full = X.gather_cols(vh_loc.t().contiguous()).t()   # [ell, d]
    idx = vh_full.abs().argmax(dim=1)
    signs = torch.sign(
        vh_full[torch.arange(vh_full.shape[0], device=vh_full.device), idx])
    signs[signs == 0] = 1.0
    vh_full = vh_full * signs[:, None]
    u_b = u_b * signs[None, :]
    u = None
    if compute_u:
        u = (Q.to(torch.float64) @ u_b[:, :k]).to(X.dtype)
    return u, s[:k].to(X.dtype), vh_full[:k].to(X.dtype)
