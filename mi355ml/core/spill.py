"""Host-pinned spill tier for > HBM datasets (SURVEY §2.5 memory mgmt).

The reference spills worker memory to disk through zict at 70% pressure;
the MI355X analog keeps the shard in PINNED host memory and streams
row blocks to the device through a dedicated copy stream, double-buffered
so the H2D of block i+1 rides under the compute of block i. Estimator hot
loops (GLM value+grad, KMeans Lloyd) accept a SpilledDistArray and
accumulate their per-shard partials over the streamed blocks — same
collectives, same numerics.
"""

from __future__ import annotations

from typing import Iterator, Optional, Tuple

import torch

from .. import config
from ..parallel import comm as comm_mod
from .array import DistArray


class SpilledDistArray(DistArray):
    """Row shard resident in (pinned) host memory; device blocks streamed."""

    def __init__(self, local: torch.Tensor, global_rows: int, row_offset: int,
                 comm=None, block_rows: Optional[int] = None):
        assert local.device.type == "cpu"
        super().__init__(local, global_rows, row_offset, comm)
        self.block_rows = block_rows or int(config.get("array.block_rows"))
        self._copy_stream = (torch.cuda.Stream(self.comm.device)
                             if self.comm.device.type == "cuda" else None)

    @property
    def device(self) -> torch.device:  # compute device, not storage
        return self.comm.device

    # -- streamed device blocks ------------------------------------------
    def device_blocks(self) -> Iterator[Tuple[int, torch.Tensor]]:
        """Yield (row_offset_local, device_block); double-buffered H2D."""
        dev = self.comm.device
        m = self.local.shape[0]
        bs = self.block_rows
        if dev.type != "cuda":
            for lo in range(0, m, bs):
                yield lo, self.local[lo:lo + bs]
            return
        starts = list(range(0, m, bs))
        bufs = [None, None]
        evts = [torch.cuda.Event(), torch.cuda.Event()]

        def _issue(i: int, slot: int):
            lo = starts[i]
            blk = self.local[lo:lo + bs]
            with torch.cuda.stream(self._copy_stream):
                bufs[slot] = blk.to(dev, non_blocking=True)
                evts[slot].record(self._copy_stream)

        if starts:
            _issue(0, 0)
        for i, lo in enumerate(starts):
            slot = i & 1
            if i + 1 < len(starts):
                _issue(i + 1, slot ^ 1)
            torch.cuda.current_stream(dev).wait_event(evts[slot])
            yield lo, bufs[slot]
            # ensure compute done before the buffer slot is reused
            evts[slot].record(torch.cuda.current_stream(dev))
            self._copy_stream.wait_event(evts[slot])

    # -- reductions stream blocks ----------------------------------------
    def moments(self):
        from .. import ops
        parts_n, parts_s, parts_m2 = [], [], []
        for _, blk in self.device_blocks():
            n, s, m2 = ops.col_moments(
                blk if blk.dim() == 2 else blk[:, None])
            parts_n.append(n.reshape(1))
            parts_s.append(s)
            parts_m2.append(m2)
        from ..ops import merge_moments
        ns = torch.cat(parts_n)
        n_l, s_l, m2_l = merge_moments(ns, torch.stack(parts_s),
                                       torch.stack(parts_m2))
        if self.comm.world_size > 1:
            f = s_l.shape[0]
            gathered = self.comm.all_gather(
                torch.cat([n_l[None], s_l, m2_l]).contiguous())
            ns = torch.stack([p[0] for p in gathered])
            sums = torch.stack([p[1:1 + f] for p in gathered])
            m2s = torch.stack([p[1 + f:] for p in gathered])
            n_l, s_l, m2_l = merge_moments(ns, sums, m2s)
        nn = float(n_l)
        mean = s_l / max(nn, 1.0)
        return nn, self._vec(mean), self._vec(m2_l)

    def min(self, axis: int = 0):
        from .. import ops
        mns = []
        for _, blk in self.device_blocks():
            mn, _ = ops.col_minmax(blk if blk.dim() == 2 else blk[:, None])
            mns.append(mn)
        out = torch.stack(mns).amin(dim=0)
        if self.comm.world_size > 1:
            self.comm.all_reduce(out, op="min")
        return self._vec(out)

    def max(self, axis: int = 0):
        from .. import ops
        mxs = []
        for _, blk in self.device_blocks():
            _, mx = ops.col_minmax(blk if blk.dim() == 2 else blk[:, None])
            mxs.append(mx)
        out = torch.stack(mxs).amax(dim=0)
        if self.comm.world_size > 1:
            self.comm.all_reduce(out, op="max")
        return self._vec(out)


    # -- transforms stream through the device ----------------------------
    def map_local(self, fn):
        """Apply fn block-by-block THROUGH the device (H2D -> fn -> D2H),
        result pinned host again — scalers/transforms on out-of-HBM data
        run on the GPU instead of host torch."""
        outs = []
        for _, blk in self.device_blocks():
            res = fn(blk)
            outs.append(res.cpu())
        local = (torch.cat(outs, dim=0) if outs else
                 fn(self.local[:0]).cpu())
        if self.comm.device.type == "cuda":
            local = local.pin_memory()
        return SpilledDistArray(local, self.global_rows, self.row_offset,
                                self.comm, self.block_rows)


def spill_rows(full: torch.Tensor, comm=None,
               block_rows: Optional[int] = None,
               pin: bool = True) -> SpilledDistArray:
    """Shard rows across ranks, keeping each shard in pinned host memory."""
    comm = comm if comm is not None else comm_mod.default()
    lo, hi = comm.shard_bounds(full.shape[0])
    local = full[lo:hi].contiguous().cpu()
    if pin and comm.device.type == "cuda":
        local = local.pin_memory()
    return SpilledDistArray(local, full.shape[0], lo, comm, block_rows)


def to_spilled(X: DistArray, block_rows: Optional[int] = None,
               pin: bool = True) -> SpilledDistArray:
    local = X.local.cpu()
    if pin and X.comm.device.type == "cuda":
        local = local.pin_memory()
    return SpilledDistArray(local, X.global_rows, X.row_offset, X.comm,
                            block_rows)
