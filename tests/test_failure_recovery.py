"""End-to-end rank-failure recovery: a rank crashes mid-fit,
torchrun --max-restarts respawns the job (the nanny analog), and the fit
resumes from the solver checkpoint instead of restarting from scratch."""

import os
import subprocess
import sys
import textwrap

import numpy as np
import torch


def test_elastic_restart_resumes_from_checkpoint(tmp_path):
    ckpt = tmp_path / "km.ckpt"
    done = tmp_path / "done.txt"
    script = tmp_path / "train.py"
    script.write_text(textwrap.dedent(f"""
        import os
        import numpy as np
        import torch
        from mi355ml.core.array import from_numpy
        from mi355ml.cluster import KMeans
        from mi355ml.checkpoint import load_checkpoint
        from mi355ml.parallel import comm as comm_mod

        comm = comm_mod.init()
        restart = int(os.environ.get("TORCHELASTIC_RESTART_COUNT", "0"))

        rng = np.random.default_rng(0)
        centers = rng.uniform(-8, 8, (4, 6))
        lab = rng.integers(0, 4, 1200)
        X = (centers[lab] + rng.standard_normal((1200, 6))).astype(
            np.float32)

        class CrashingKMeans(KMeans):
            def _init_centers(self, Xd):
                c = super()._init_centers(Xd)
                return c

        km = CrashingKMeans(n_clusters=4, random_state=0, max_iter=12,
                            tol=0.0, checkpoint_path={str(ckpt)!r},
                            checkpoint_every=1)
        if restart == 0:
            # first attempt: crash rank 1 after a few iterations by
            # monkey-patching the checkpoint step
            from mi355ml import checkpoint as ck
            orig = ck.CheckpointManager.step

            def crashing_step(self, iteration, **state):
                orig(self, iteration, **state)
                if iteration >= 3 and comm.rank == 1:
                    os._exit(17)  # simulated hard rank death

            ck.CheckpointManager.step = crashing_step
        st = load_checkpoint({str(ckpt)!r})
        start = 0 if st is None else int(st["iteration"])
        km.fit(from_numpy(X))
        if comm.rank == 0:
            with open({str(done)!r}, "w") as f:
                f.write(f"restart={{restart}} resumed_from={{start}} "
                        f"n_iter={{km.n_iter_}} inertia={{km.inertia_}}")
        comm_mod.shutdown()
    """))
    from conftest import free_port
    env = dict(os.environ)
    env["CUDA_VISIBLE_DEVICES"] = ""
    env["HIP_VISIBLE_DEVICES"] = ""
    env["PYTHONPATH"] = os.getcwd()
    env["MI355ML_COMM__TIMEOUT"] = "30"  # fail fast on the dead peer
    res = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node=2", "--master-addr=127.0.0.1",
         f"--master-port={free_port()}", "--max-restarts=2",
         "--monitor-interval=1", str(script)],
        env=env, capture_output=True, text=True, timeout=420)
    assert done.exists(), (res.stdout[-2000:], res.stderr[-2000:])
    content = done.read_text()
    # the job finished on a RESPAWNED attempt, resuming past iteration 3
    assert "restart=1" in content or "restart=2" in content, content
    resumed = int(content.split("resumed_from=")[1].split()[0])
    assert resumed >= 3, content
