import os
import socket

import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD GPU (run with -m gpu)")


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
    p = s.getsockname()[1]
    s.close()
    return p


def run_distributed(fn, world_size, args=(), backend="gloo", timeout=120):
    """Spawn world_size processes running fn(rank, world_size, *args).

    Results are returned as a list indexed by rank (whatever fn returns,
    via a multiprocessing queue). Raises on any rank failure.
    """
    import torch.multiprocessing as mp

    port = free_port()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()

    import queue as _q
    import time as _t

    ps = []
    for rank in range(world_size):
        p = ctx.Process(target=_dist_entry,
                        args=(fn, rank, world_size, port, backend, q, args))
        p.start()
        ps.append(p)
    results = {}
    deadline = _t.time() + timeout
    try:
        for _ in range(world_size):
            while True:
                try:
                    rank, ok, payload = q.get(timeout=5)
                    break
                except _q.Empty:
                    dead = [p for p in ps if not p.is_alive()
                            and p.exitcode not in (0, None)]
                    if dead:
                        raise RuntimeError(
                            f"worker died with exitcode "
                            f"{[p.exitcode for p in dead]}")
                    if _t.time() > deadline:
                        raise RuntimeError(
                            f"distributed test hung (> {timeout}s)")
            if not ok:
                raise RuntimeError(f"rank {rank} failed:\n{payload}")
            results[rank] = payload
        for p in ps:
            p.join(30)
    finally:
        for p in ps:
            if p.is_alive():
                p.terminate()
    return [results[r] for r in range(world_size)]


def _dist_entry(fn, rank, world, port, backend, q, args):
    import traceback

    import torch.distributed as dist

    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        dist.init_process_group(backend, rank=rank, world_size=world)
        out = fn(rank, world, *args)
        q.put((rank, True, out))
    except Exception:
        q.put((rank, False, traceback.format_exc()))
