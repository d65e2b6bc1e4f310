"""The driver contract: bench.py under torchrun, one rank per device,
prints ONE JSON line with the whole-job aggregate. Exercised here on
gloo/CPU with a tiny synthetic size so the 8-GPU round-end run can't be
the first time the distributed path executes (VERDICT r1 #1c covers
world=4 as well as world=2)."""

import json
import os
import subprocess
import sys

import pytest

from conftest import free_port

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run_bench(nproc, extra, timeout=420):
    env = dict(os.environ)
    env["CUDA_VISIBLE_DEVICES"] = ""
    env["HIP_VISIBLE_DEVICES"] = ""
    if nproc == 1:
        cmd = [sys.executable, "bench.py"]
    else:
        cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
               f"--nproc-per-node={nproc}", "--master-addr=127.0.0.1",
               f"--master-port={free_port()}", "bench.py"]
    res = subprocess.run(cmd + extra, env=env, capture_output=True,
                         text=True, timeout=timeout, cwd=REPO)
    assert res.returncode == 0, res.stderr[-2000:]
    line = [l for l in res.stdout.strip().splitlines()
            if l.startswith("{")][-1]
    return json.loads(line)


@pytest.mark.parametrize("world", [2, 4, 8])
def test_bench_distributed_contract(world):
    d = _run_bench(world, ["--steps", "2", "--warmup", "1",
                           "--rows", "100000"])
    assert d["n_gpus"] == world
    assert d["steps"] == 2 and d["warmup"] == 1
    # strong scaling: the GLOBAL problem is the named shape at every N
    assert d["scaling"] == "strong"
    assert d["higher_is_better"] is True
    assert d["config"]["parallelism"] == f"dp{world}"
    assert d["config"]["global_batch"] == 100000
    assert d["config"]["rows_per_gpu"] == 100000 // world
    # whole-job aggregate: value = global_rows / step_time
    assert d["value"] > 0 and d["ms_per_step"] > 0
    expected = 100000 / (d["ms_per_step"] / 1e3)
    assert abs(d["value"] - expected) / expected < 0.05
    # per-collective counters travel with the report
    assert d["config"]["counters"].get("comm.all_reduce.calls", 0) > 0


def test_bench_single_process_default():
    d = _run_bench(1, ["--steps", "2", "--warmup", "1",
                       "--rows", "100000"], timeout=300)
    assert d["n_gpus"] == 1
    assert d["config"]["global_batch"] == 100000


@pytest.mark.parametrize("cfg,world", [("2", 2), ("3", 2), ("4", 2),
                                       ("5", 2)])
def test_bench_config_modes_world2(cfg, world):
    d = _run_bench(world, ["--config", cfg, "--steps", "2",
                           "--warmup", "1"])
    assert d["n_gpus"] == world
    assert d["value"] > 0 and d["ms_per_step"] > 0
    assert d["higher_is_better"] is True
    assert "config" in d and "counters" in d["config"]


def test_one_step_math_matches_reference():
    """The bench's fused step must compute a REAL Lloyd update + a real
    gradient step (guards against accidentally benchmarking skipped
    work)."""
    import numpy as np
    import torch

    import bench
    from mi355ml.parallel import comm as comm_mod

    comm = comm_mod.default()
    rng = np.random.default_rng(3)
    x = torch.from_numpy(rng.standard_normal((500, 8)).astype(np.float32))
    y = torch.from_numpy((rng.uniform(size=500) < 0.5).astype(np.float32))
    c0 = torch.from_numpy(rng.standard_normal((4, 8)).astype(np.float32))
    from mi355ml.core.array import DistArray
    X = DistArray(x, 500, 0, comm)
    Y = DistArray(y, 500, 0, comm)
    w0 = torch.zeros(8)
    buf = torch.empty(2 + 8 + 4 * 8 + 4 + 1, dtype=torch.float64)
    w1, b1, c1, loss, inertia = bench.one_step(X, Y, w0.clone(), 0.0,
                                               c0.clone(), 1.0, comm, buf)
    # reference Lloyd update
    d2 = ((x[:, None, :] - c0[None]) ** 2).sum(-1)
    lab = d2.argmin(1)
    c_ref = torch.stack([
        x[lab == j].mean(0) if (lab == j).any() else torch.zeros(8)
        for j in range(4)])
    torch.testing.assert_close(c1, c_ref, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(
        torch.tensor(inertia, dtype=torch.float64),
        d2.min(1).values.double().sum(), rtol=1e-5, atol=1e-2)
    # reference logistic gradient step (lr/n scaling)
    z = x @ w0
    r = torch.sigmoid(z) - y
    gw = x.t() @ r
    torch.testing.assert_close(w1.double(), (w0 - gw / 500).double(),
                               rtol=1e-5, atol=1e-6)
    ref_loss = (torch.nn.functional.softplus(z) - y * z).double().sum()
    torch.testing.assert_close(torch.tensor(loss, dtype=torch.float64),
                               ref_loss, rtol=1e-5, atol=1e-3)
