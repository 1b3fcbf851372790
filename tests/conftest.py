import os
import socket
import sys

import pytest
import torch

# make the repo root importable regardless of pytest invocation dir
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs a ROCm GPU (run on MI355X box)")


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


def run_dist(fn, world_size: int, args=(), backend: str = "gloo", timeout: float = 180.0):
    """Spawn `world_size` processes, init gloo on 127.0.0.1, run fn(rank, *args).

    free_port() is racy (the port is released before the children bind, so a
    concurrent test worker can steal it); a rendezvous-flavored failure gets
    ONE retry on a fresh port — a genuine test failure still fails twice."""
    import torch.multiprocessing as mp

    last_err = None
    for attempt in range(2):
        port = free_port()
        ctx = mp.get_context("spawn")
        procs = []
        for rank in range(world_size):
            p = ctx.Process(
                target=_dist_entry, args=(fn, rank, world_size, port, backend, args)
            )
            p.start()
            procs.append(p)
        for p in procs:
            p.join(timeout)
        failed = None
        for rank, p in enumerate(procs):
            if p.is_alive():
                p.terminate()
                failed = TimeoutError(f"rank {rank} timed out")
            elif p.exitcode != 0 and failed is None:
                failed = AssertionError(f"rank {rank} exited with {p.exitcode}")
        if failed is None:
            return
        last_err = failed
    raise last_err


def _dist_entry(fn, rank, world, port, backend, args):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group(backend, rank=rank, world_size=world)
    try:
        fn(rank, *args)
    finally:
        dist.destroy_process_group()
