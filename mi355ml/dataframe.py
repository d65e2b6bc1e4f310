"""Partitioned-dataframe helpers (the used subset of the reference's
dask.dataframe layer — SURVEY §2.3 notes only df-preprocessing is needed;
Categorizer/DummyEncoder live in preprocessing._dataframe).

Frames are pandas DataFrames partitioned over ranks in a DistList. The
groupby aggregation follows the reference's ACA (apply-concat-apply)
groupby: per-partition partial aggregates, ONE ragged object all-gather,
and a deterministic combine — mean carries (sum, count) partials so the
result is exact, never an average of averages.
"""

from __future__ import annotations

from typing import Callable, List, Optional, Union

import numpy as np
import pandas as pd

from .core.bag import DistList


def from_pandas(df: pd.DataFrame, npartitions: Optional[int] = None
                ) -> DistList:
    """Split a DataFrame into row partitions sharded over ranks."""
    from .parallel import comm as comm_mod
    comm = comm_mod.default()
    nparts = npartitions or comm.world_size
    bounds = np.linspace(0, len(df), nparts + 1).astype(int)
    parts = [df.iloc[bounds[i]:bounds[i + 1]] for i in range(nparts)]
    return DistList.from_global(parts)


def map_partitions(fn: Callable, frames: DistList) -> DistList:
    """Apply fn to every partition (dask map_partitions analog)."""
    return frames.map(fn)


def concat(frames: DistList) -> pd.DataFrame:
    """Materialize the full frame on every rank."""
    parts = frames.gather()
    return pd.concat(parts) if parts else pd.DataFrame()


def groupby_agg(frames: DistList, by: Union[str, List[str]],
                agg: str = "sum",
                columns: Optional[List[str]] = None) -> pd.DataFrame:
    """Distributed groupby aggregate: agg in {sum, count, mean, min, max}.

    Returns the aggregated frame, replicated on every rank.
    """
    if agg not in ("sum", "count", "mean", "min", "max"):
        raise ValueError(f"unsupported agg {agg!r}")

    def _partial(part: pd.DataFrame):
        if part is None or not len(part):
            return None
        sub = part if columns is None else part[
            ([by] if isinstance(by, str) else list(by)) + list(columns)]
        g = sub.groupby(by)
        if agg == "mean":  # carry (sum, count) — never average averages
            return {"sum": g.sum(numeric_only=True),
                    "count": g.count()}
        if agg == "count":
            return {"count": g.count()}
        return {agg: getattr(g, agg)(numeric_only=(agg == "sum"))}

    local = [_partial(p) for p in frames.local]
    gathered = frames.comm.all_gather_object(local)
    partials = [p for rank_parts in gathered for p in rank_parts
                if p is not None]
    if not partials:
        return pd.DataFrame()

    def _combine(key: str, how: str) -> pd.DataFrame:
        cat = pd.concat([p[key] for p in partials])
        g = cat.groupby(level=list(range(cat.index.nlevels)))
        return getattr(g, how)()

    if agg == "mean":
        s = _combine("sum", "sum")
        c = _combine("count", "sum")
        return s / c
    if agg in ("sum", "count"):
        return _combine(agg if agg != "count" else "count", "sum")
    return _combine(agg, agg)  # min of mins / max of maxes


def _key_frame(part: pd.DataFrame, by) -> pd.Series:
    cols = [by] if isinstance(by, str) else list(by)
    # deterministic cross-process hash (fixed-key siphash)
    return pd.util.hash_pandas_object(part[cols], index=False)


def shuffle(frames: DistList, on: Union[str, List[str]]) -> DistList:
    """Hash-partition rows by key so all rows of a key land on ONE rank
    (the reference's rearrange_by_column task shuffle,
    dask/dataframe/shuffle.py:443). Enables exact groupby-apply and joins.

    Exchange is true point-to-point (the task-shuffle data plane): each
    (src, dst) pair moves exactly its pickled slice with one isend/irecv
    over the direct link — not a W-fold object all-gather."""
    import pickle

    import torch
    import torch.distributed as dist

    comm = frames.comm
    W = comm.world_size
    if W == 1:
        parts = [p for p in frames.local if p is not None and len(p)]
        # one partition so every key group is contiguous for groupby-apply
        return DistList([pd.concat(parts) if parts else pd.DataFrame()],
                        comm)
    outgoing: List[List[pd.DataFrame]] = [[] for _ in range(W)]
    for part in frames.local:
        if part is None or not len(part):
            continue
        dest = (_key_frame(part, on) % W).to_numpy()
        for r in range(W):
            sel = part[dest == r]
            if len(sel):
                outgoing[r].append(sel)
    template = next((p.iloc[:0] for p in frames.local
                     if p is not None and p.shape[1]), None)
    blobs = [pickle.dumps(pd.concat(o)) if o else b"" for o in outgoing]
    # size matrix via one small all-gather, then batched P2P of the blobs
    sizes_local = torch.tensor([len(b) for b in blobs], dtype=torch.int64)
    sizes = [s.cpu() for s in comm.all_gather(sizes_local.to(comm.device)
                                              if comm.device.type == "cuda"
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
    mine = []
    if blobs[comm.rank]:
        mine.append(pickle.loads(blobs[comm.rank]))
    for r in sorted(recv_bufs):
        mine.append(pickle.loads(recv_bufs[r].numpy().tobytes()))
    # template exchange keeps empty ranks schema-correct for merges
    templates = comm.all_gather_object(template)
    if mine:
        return DistList([pd.concat(mine)], comm)
    empty = next((t for t in templates if t is not None), pd.DataFrame())
    return DistList([empty], comm)


def groupby_apply(frames: DistList, by: Union[str, List[str]],
                  fn: Callable) -> DistList:
    """Exact groupby-apply: shuffle by key, then pandas groupby per
    partition (every group is wholly on one rank after the shuffle)."""
    shuffled = shuffle(frames, by)

    def _apply(part: pd.DataFrame):
        if part is None or not len(part):
            return part
        # explicit per-group loop: fn sees the FULL sub-frame including the
        # key columns (dask groupby.apply semantics), independent of the
        # pandas include_groups deprecation cycle
        outs = [fn(sub) for _, sub in part.groupby(by, group_keys=False,
                                                   sort=True)]
        return pd.concat(outs) if outs else part.iloc[:0]

    return shuffled.map(_apply)


def merge(left: DistList, right: DistList, on: Union[str, List[str]],
          how: str = "inner") -> DistList:
    """Distributed join: co-shuffle both frames by key, then per-rank
    pandas merge (dask.dataframe.multi hash-join analog)."""
    ls = shuffle(left, on)
    rs = shuffle(right, on)
    lp = ls.local[0] if ls.local else pd.DataFrame()
    rp = rs.local[0] if rs.local else pd.DataFrame()
    if not len(lp) and not len(rp):
        return DistList([pd.DataFrame()], left.comm)
    return DistList([lp.merge(rp, on=on, how=how)], left.comm)


def value_counts(frames: DistList, column: str) -> pd.Series:
    def _vc(part):
        return part[column].value_counts() if len(part) else None

    local = [_vc(p) for p in frames.local]
    gathered = frames.comm.all_gather_object(local)
    parts = [p for rp in gathered for p in rp if p is not None]
    if not parts:
        return pd.Series(dtype=np.int64)
    return pd.concat(parts).groupby(level=0).sum().sort_values(
        ascending=False)
