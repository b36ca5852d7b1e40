import os
import socket
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: needs an MI355X (run with -m gpu on a GPU box)")


def pytest_collection_modifyitems(config, items):
    import torch
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU in this container")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


def free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def run_distributed(worker, world_size: int, args=(), timeout: float = 180.0):
    """Spawn `world_size` processes; each runs
    worker(rank, world_size, port, tmpdir, *args) after gloo init is set up
    by the worker itself (via tests.dist_utils.init_worker)."""
    import tempfile

    import torch.multiprocessing as mp

    import time

    last_exc = None
    for attempt in range(2):      # one retry absorbs port/rendezvous flakes
        port = free_port()
        tmpdir = tempfile.mkdtemp(prefix="acco_test_")
        try:
            ctx = mp.spawn(worker, args=(world_size, port, tmpdir) + tuple(args),
                           nprocs=world_size, join=False)
            deadline = time.time() + timeout
            # ctx.join(t) returns False whenever *some* process is still
            # alive after one wait round — poll until the deadline.
            while not ctx.join(timeout=5):
                if time.time() > deadline:
                    for p in ctx.processes:
                        p.terminate()
                    raise RuntimeError("distributed test timed out")
            return tmpdir
        except Exception as e:   # ProcessRaisedException is not a RuntimeError
            last_exc = e
    raise last_exc
