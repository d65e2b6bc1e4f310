"""Property-based tests (hypothesis) for the merge math the distributed
reductions rely on: Chan moment merge, percentile curve merge, t-digest
build/merge — invariants must hold for ANY partition of the rows, because
rank count and block boundaries are deployment details."""

import numpy as np
import torch
from hypothesis import given, settings, strategies as st

from mi355ml.ops import merge_moments


def _split(arr, cuts):
    """Partition rows of arr at the given sorted cut fractions."""
    n = arr.shape[0]
    idx = sorted({int(c * n) for c in cuts} | {0, n})
    return [arr[a:b] for a, b in zip(idx[:-1], idx[1:])]


@settings(max_examples=40, deadline=None, derandomize=True)
@given(
    n=st.integers(20, 400),
    f=st.integers(1, 5),
    seed=st.integers(0, 10_000),
    cuts=st.lists(st.floats(0.01, 0.99), min_size=1, max_size=6),
    scale=st.floats(1e-3, 1e3),
)
def test_moment_merge_partition_invariant(n, f, seed, cuts, scale):
    rng = np.random.default_rng(seed)
    X = (rng.standard_normal((n, f)) * scale).astype(np.float64)
    parts = [p for p in _split(X, cuts) if p.shape[0] > 0]
    ns, sums, m2s = [], [], []
    for p in parts:
        t = torch.from_numpy(p)
        ns.append(float(p.shape[0]))
        sums.append(t.sum(dim=0))
        mu = t.mean(dim=0)
        m2s.append(((t - mu) ** 2).sum(dim=0))
    ntot, s, m2 = merge_moments(torch.tensor(ns, dtype=torch.float64),
                                torch.stack(sums), torch.stack(m2s))
    assert float(ntot) == n
    np.testing.assert_allclose(s.numpy() / n, X.mean(axis=0),
                               rtol=1e-10, atol=1e-12 * scale)
    np.testing.assert_allclose(m2.numpy() / n, X.var(axis=0),
                               rtol=1e-8, atol=1e-10 * scale * scale)


@settings(max_examples=30, deadline=None, derandomize=True)
@given(
    n=st.integers(50, 500),
    seed=st.integers(0, 10_000),
    cuts=st.lists(st.floats(0.05, 0.95), min_size=1, max_size=5),
    qs=st.lists(st.floats(1.0, 99.0), min_size=1, max_size=6, unique=True),
)
def test_percentile_merge_bounds_and_monotone(n, seed, cuts, qs):
    from mi355ml.ops.percentile import merge_percentiles
    rng = np.random.default_rng(seed)
    x = rng.standard_normal(n).astype(np.float64)
    qs = sorted(qs)
    q_ext = torch.tensor(sorted({0.0, 100.0} | set(qs)), dtype=torch.float64)
    curves, ns = [], []
    for p in _split(x[:, None], cuts):
        if p.shape[0] == 0:
            continue
        t = torch.from_numpy(p)
        curves.append(torch.quantile(t, q_ext / 100.0, dim=0))
        ns.append(float(p.shape[0]))
    merged = merge_percentiles(qs, torch.stack(curves),
                               torch.tensor(ns, dtype=torch.float64), q_ext)
    vals = merged[:, 0].numpy()
    # invariants of the weighted-CDF merge:
    # 1) monotone in q and inside the data range
    assert (np.diff(vals) >= -1e-12).all()
    assert vals.min() >= x.min() - 1e-9 and vals.max() <= x.max() + 1e-9
    # 2) each merged q-quantile is bracketed by the per-part q-quantiles
    #    (a part with everything below/above pulls the merge toward it,
    #    but never past the extreme part)
    per_part = np.stack([c.numpy()[:, 0] for c in curves])  # [parts, nq_ext]
    qe = q_ext.numpy()
    for q, v in zip(qs, vals):
        j = int(np.where(qe == q)[0][0])
        lo = per_part[:, j].min()
        hi = per_part[:, j].max()
        assert lo - 1e-9 <= v <= hi + 1e-9
    # 3) single partition -> exact quantiles
    if len(curves) == 1:
        exact = np.quantile(x, np.asarray(qs) / 100.0)
        np.testing.assert_allclose(vals, exact, rtol=1e-9, atol=1e-9)


@settings(max_examples=25, deadline=None, derandomize=True)
@given(
    n=st.integers(200, 3000),
    seed=st.integers(0, 10_000),
    nparts=st.integers(1, 5),
)
def test_tdigest_merge_rank_accuracy(n, seed, nparts):
    from mi355ml.ops.percentile import (_tdigest_build, _tdigest_merge,
                                        _tdigest_query)
    rng = np.random.default_rng(seed)
    x = rng.standard_normal(n).astype(np.float64)
    bounds = np.linspace(0, n, nparts + 1).astype(int)
    ms, ws = [], []
    for a, b in zip(bounds[:-1], bounds[1:]):
        part = torch.from_numpy(x[a:b, None])
        if part.shape[0] == 0:
            continue
        m, w = _tdigest_build(part, torch.ones_like(part), 500)
        ms.append(m)
        ws.append(w)
    m, w = _tdigest_merge(torch.cat(ms, dim=1), torch.cat(ws, dim=1), 500)
    assert float(w.sum()) == n  # no mass lost in the merge
    qs = [5.0, 25.0, 50.0, 75.0, 95.0]
    mins = torch.tensor([x.min()], dtype=torch.float64)
    maxs = torch.tensor([x.max()], dtype=torch.float64)
    est = _tdigest_query(m, w, mins, maxs, qs)[:, 0].numpy()
    order = np.sort(x)
    for q, v in zip(qs, est):
        r = np.searchsorted(order, v) / n * 100
        assert abs(r - q) < 3.0  # t-digest rank error bound (delta=500)


@settings(max_examples=20, deadline=None, derandomize=True)
@given(seed=st.integers(0, 2**31 - 1))
def test_shard_seed_streams_distinct(seed):
    from mi355ml.core.random import shard_seed
    seen = set()
    for shard in range(8):
        for stream in range(4):
            seen.add(shard_seed(seed, shard, stream))
    assert len(seen) == 32  # no collisions across (shard, stream)
