import os
import socket

import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU (run via gpurun)")


def _free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _dist_entry(rank, world, port, fn, args, backend, results):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist
    dist.init_process_group(backend, rank=rank, world_size=world)
    try:
        out = fn(rank, world, *args)
        if results is not None:
            results[rank] = out
    finally:
        dist.destroy_process_group()


def run_distributed(fn, world=2, args=(), backend="gloo", timeout=180):
    """Runs fn(rank, world, *args) in `world` processes with a gloo group.

    Mirrors the reference's real-multi-process test harness (SURVEY.md §4:
    tests run under `horovodrun -np N`, no fake backend).
    """
    import torch.multiprocessing as mp
    port = _free_port()
    ctx = mp.get_context("spawn")
    mgr = ctx.Manager()
    results = mgr.dict()
    procs = []
    for r in range(world):
        p = ctx.Process(target=_dist_entry, args=(r, world, port, fn, args, backend, results))
        p.start()
        procs.append(p)
    for p in procs:
        p.join(timeout)
    hung = [r for r, p in enumerate(procs) if p.is_alive()]
    if hung:
        for p in procs:           # terminate ALL stragglers, not just the first
            if p.is_alive():
                p.terminate()
        for p in procs:
            p.join(10)
        raise RuntimeError(f"ranks {hung} timed out")
    for r, p in enumerate(procs):
        assert p.exitcode == 0, f"rank {r} exited with {p.exitcode}"
    return [results.get(r) for r in range(world)]


@pytest.fixture
def seed():
    torch.manual_seed(1234)
    return 1234
