import os
import socket
import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent
if str(REPO) not in sys.path:
    sys.path.insert(0, str(REPO))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU (run with -m gpu)")


def pytest_collection_modifyitems(config, items):
    import torch

    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this environment")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


def free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def run_distributed(fn, world_size: int, args=(), timeout: float = 120.0):
    """Spawn `world_size` processes running fn(rank, world_size, *args) with
    torch.distributed env (gloo over 127.0.0.1) prepared."""
    import multiprocessing as mp

    port = free_port()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = []
    for rank in range(world_size):
        p = ctx.Process(
            target=_dist_entry, args=(fn, rank, world_size, port, q, args)
        )
        p.start()
        procs.append(p)
    results = {}
    errs = []
    for _ in range(world_size):
        kind, rank, payload = q.get(timeout=timeout)
        if kind == "ok":
            results[rank] = payload
        else:
            errs.append((rank, payload))
    for p in procs:
        p.join(timeout=30)
        if p.is_alive():
            p.terminate()
    if errs:
        raise AssertionError(f"distributed workers failed: {errs}")
    return [results[r] for r in range(world_size)]


def _dist_entry(fn, rank, world_size, port, q, args):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    try:
        out = fn(rank, world_size, *args)
        q.put(("ok", rank, out))
    except Exception as e:  # noqa: BLE001
        import traceback

        q.put(("err", rank, f"{e}\n{traceback.format_exc()}"))
