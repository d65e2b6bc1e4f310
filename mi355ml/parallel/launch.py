"""mi355ml launcher — the single-command analog of the reference's
dask-scheduler/dask-worker CLIs: spawn one rank process per
GPU over RCCL with torchrun-compatible env.

    python -m mi355ml.parallel.launch --gpus 8 my_script.py [args...]
"""

from __future__ import annotations

import argparse
import os
import sys


def main(argv=None):
    ap = argparse.ArgumentParser(prog="mi355ml-launch")
    ap.add_argument("--gpus", "-n", type=int, default=None,
                    help="ranks (default: visible GPU count, else 1)")
    ap.add_argument("--master-port", type=int, default=29531)
    ap.add_argument("--max-restarts", type=int, default=0,
                    help="elastic rank-respawn attempts on failure (the "
                         "reference's nanny auto-restart analog; solvers "
                         "resume from their checkpoints — SURVEY §5.3)")
    ap.add_argument("script")
    ap.add_argument("script_args", nargs=argparse.REMAINDER)
    args = ap.parse_args(argv)

    n = args.gpus
    if n is None:
        try:
            import torch
            n = max(1, torch.cuda.device_count())
        except Exception:
            n = 1
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    from torch.distributed.run import main as torchrun
    extra = []
    if args.max_restarts > 0:
        # fast failure detection: the default 5 s monitor interval delays
        # the respawn well past the surviving ranks' comm timeout
        extra.append("--monitor-interval=1")
    torchrun([
        "--nnodes=1",
        f"--nproc-per-node={n}",
        "--master-addr=127.0.0.1",
        f"--master-port={args.master_port}",
        f"--max-restarts={args.max_restarts}",
        *extra,
        args.script,
        *args.script_args,
    ])


if __name__ == "__main__":
    main()
