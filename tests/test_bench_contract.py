"""The driver contract: bench.py under torchrun, one rank per device,
prints ONE JSON line with the whole-job aggregate. Exercised here on
gloo/CPU with a tiny synthetic size so the 8-GPU round-end run can't be
the first time the distributed path executes."""

import json
import os
import subprocess
import sys

from conftest import free_port


def test_bench_distributed_contract():
    env = dict(os.environ)
    env["CUDA_VISIBLE_DEVICES"] = ""
    env["HIP_VISIBLE_DEVICES"] = ""
    res = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node=2", "--master-addr=127.0.0.1",
         f"--master-port={free_port()}", "bench.py", "--steps", "2",
         "--warmup", "1", "--rows-per-gpu", "100000"],
        env=env, capture_output=True, text=True, timeout=420,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert res.returncode == 0, res.stderr[-2000:]
    line = [l for l in res.stdout.strip().splitlines()
            if l.startswith("{")][-1]
    d = json.loads(line)
    assert d["n_gpus"] == 2
    assert d["steps"] == 2 and d["warmup"] == 1
    assert d["scaling"] == "weak"
    assert d["higher_is_better"] is True
    assert d["config"]["parallelism"] == "dp2"
    # whole-job aggregate: value = rows_per_gpu * n_gpus / step_time
    assert d["value"] > 0 and d["ms_per_step"] > 0
    expected = 100000 * 2 / (d["ms_per_step"] / 1e3)
    assert abs(d["value"] - expected) / expected < 0.05


def test_bench_single_process_default():
    env = dict(os.environ)
    env["CUDA_VISIBLE_DEVICES"] = ""
    env["HIP_VISIBLE_DEVICES"] = ""
    res = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
         "--rows-per-gpu", "100000"],
        env=env, capture_output=True, text=True, timeout=300,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert res.returncode == 0, res.stderr[-2000:]
    d = json.loads([l for l in res.stdout.strip().splitlines()
                    if l.startswith("{")][-1])
    assert d["n_gpus"] == 1
