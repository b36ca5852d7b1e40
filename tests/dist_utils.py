"""Helpers for multi-process gloo tests (world_size>1 on CPU)."""

import datetime
import os

import torch.distributed as dist


def init_worker(rank: int, world_size: int, port: int) -> None:
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    # torchrun-style env so DistContext.detect_topology sees the real rank
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    dist.init_process_group("gloo", rank=rank, world_size=world_size,
                            timeout=datetime.timedelta(seconds=120))


def teardown_worker() -> None:
    if dist.is_initialized():
        dist.destroy_process_group()
