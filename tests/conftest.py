import os
import socket

import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run with -m gpu)")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


def free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _mp_entry(rank, world, port, fn, args):
    # gloo + CPU tensors even on a GPU box: multi-rank CPU tests must behave
    # identically everywhere (gloo does not support every cuda collective)
    os.environ["CUDA_VISIBLE_DEVICES"] = ""
    os.environ["HIP_VISIBLE_DEVICES"] = ""
    os.environ["ROCR_VISIBLE_DEVICES"] = ""
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import mi355ml.parallel.comm as comm
    c = comm.init(backend="gloo")
    try:
        fn(rank, world, *args)
        # a rank that finishes early must not exit (killing the rank-0
        # TCPStore) while peers still use control-plane primitives
        c.barrier()
    finally:
        comm.shutdown()


def run_multiprocess(fn, world: int = 2, args: tuple = (),
                     timeout: float = 300.0):
    """Run fn(rank, world, *args) on `world` gloo-connected processes
    (the gen_cluster analog — SURVEY §4 tier 2). Bounded join: a
    deadlocked rank fails the test instead of hanging the whole suite."""
    import time

    import torch.multiprocessing as mp
    port = free_port()
    ctx = mp.start_processes(_mp_entry, args=(world, port, fn, args),
                             nprocs=world, join=False,
                             start_method="spawn")
    deadline = time.monotonic() + timeout
    while not ctx.join(timeout=5):
        if time.monotonic() > deadline:
            for p in ctx.processes:
                if p.is_alive():
                    p.terminate()
            raise TimeoutError(
                f"multiprocess test exceeded {timeout:.0f}s "
                f"(world={world}); ranks terminated")


@pytest.fixture
def mp_runner():
    return run_multiprocess
