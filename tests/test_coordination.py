"""Coordination primitives (Event/Lock/Variable/Queue/Semaphore) —
single-process and gloo world-2."""

import numpy as np

from mi355ml.parallel.coordination import (Event, Lock, Queue, Semaphore,
                                           Variable)


def test_local_primitives():
    v = Variable("v1")
    v.set({"a": 1})
    assert v.get() == {"a": 1}
    e = Event("e1")
    assert not e.is_set()
    e.set()
    assert e.is_set()
    e.wait(timeout=1)
    with Lock("l1"):
        pass
    q = Queue("q1")
    q.put(10)
    q.put("x")
    assert q.qsize() == 2
    assert q.get(timeout=1) == 10
    assert q.get(timeout=1) == "x"
    s = Semaphore(2, "s1")
    assert s.acquire(timeout=1)
    assert s.acquire(timeout=1)
    assert not s.acquire(timeout=0.1)
    s.release()
    assert s.acquire(timeout=1)
    s.release(); s.release()


def _dist_body(rank, world):
    v = Variable("dv")
    e = Event("de")
    q = Queue("dq")
    if rank == 0:
        v.set(42)
        q.put("from0")
        e.set()
    else:
        e.wait(timeout=30)
        assert v.get() == 42
        assert q.get(timeout=30) == "from0"
    with Lock("dl"):
        val = Variable("counter")
        try:
            cur = val.get()
        except Exception:
            cur = 0
        val.set(cur + 1)


def test_dist_primitives_world2(mp_runner):
    mp_runner(_dist_body, world=2)


def test_futures_local():
    from mi355ml.parallel.futures import FuturePool
    pool = FuturePool("t1")
    futs = pool.map(lambda x: x * x, range(5))
    assert pool.gather(futs) == [0, 1, 4, 9, 16]
    f = pool.submit(lambda: 1 / 0)
    pool.run()
    import pytest as _pytest
    with _pytest.raises(RuntimeError, match="failed on rank"):
        f.result()


def _futures_body(rank, world):
    from mi355ml.parallel.futures import FuturePool
    pool = FuturePool("t2")
    futs = [pool.submit(lambda x=x: (x, __import__("os").getpid()))
            for x in range(6)]
    pool.run()
    vals = [f.result(timeout=30) for f in futs]
    assert [v[0] for v in vals] == list(range(6))
    # tasks actually ran on different processes (round-robin owners)
    assert len({v[1] for v in vals}) == world


def test_futures_world2(mp_runner):
    mp_runner(_futures_body, world=2)


def _comm_counters_world2(rank, world):
    import torch
    from mi355ml.diagnostics import COUNTERS
    from mi355ml.parallel import comm as comm_mod
    comm = comm_mod.default()
    t = torch.ones(100, dtype=torch.float64)
    comm.all_reduce(t)
    comm.all_gather(t)
    snap = COUNTERS.snapshot()
    assert snap["comm.all_reduce.calls"] >= 1
    assert snap["comm.all_reduce.bytes"] >= 800
    assert snap["comm.all_gather.calls"] >= 1


def test_comm_counters_world2(mp_runner):
    mp_runner(_comm_counters_world2, world=2, args=())


def _joblib_ranks_world2(rank, world):
    import os
    import numpy as np
    from joblib import Parallel, delayed, parallel_backend
    import mi355ml.joblib_backend as jb
    from mi355ml.parallel import comm as comm_mod
    comm_mod.default()
    jb.register()

    def task(i):
        return (i * 10, os.getpid())

    with parallel_backend("mi355ml_ranks"):
        out = Parallel()(delayed(task)(i) for i in range(8))
    vals = [v for v, _ in out]
    assert vals == [i * 10 for i in range(8)]
    # tasks really ran in TWO different processes (round-robin ownership)
    pids = {p for _, p in out}
    assert len(pids) == 2
    # every rank sees the same pids (results were broadcast from owners)
    gathered = comm_mod.default().all_gather_object(sorted(pids))
    assert gathered[0] == gathered[1]


def test_joblib_rank_fanout_world2(mp_runner):
    mp_runner(_joblib_ranks_world2, world=2, args=())
