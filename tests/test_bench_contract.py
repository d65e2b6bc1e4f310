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
