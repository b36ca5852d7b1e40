"""Distributed bootstrap: one process per GPU over RCCL.

Topology sources, in priority order (the capability of reference
trainer_base.py:135-180, which is SLURM-only):
1. torchrun-style env (RANK / LOCAL_RANK / WORLD_SIZE / MASTER_ADDR);
2. SLURM env (SLURM_PROCID / SLURM_LOCALID / SLURM_NTASKS / hostlist);
3. single-process fallback.

Backend is "nccl" (RCCL on ROCm) when a GPU is present, else "gloo" for
CPU multi-process tests.
"""

from __future__ import annotations

import datetime
import os
from dataclasses import dataclass

import torch
import torch.distributed as dist

from acco_amd.utils.hostlist import expand_hostlist


@dataclass
class DistContext:
    rank: int
    local_rank: int
    world_size: int
    n_nodes: int
    node_id: int
    device: torch.device
    backend: str
    id_run: str

    @property
    def is_main(self) -> bool:
        return self.rank == 0


def detect_topology() -> dict:
    env = os.environ
    if "RANK" in env and "WORLD_SIZE" in env:
        rank = int(env["RANK"])
        world = int(env["WORLD_SIZE"])
        local = int(env.get("LOCAL_RANK", rank))
        nnodes = int(env.get("NNODES", max(1, world // max(1, int(env.get("LOCAL_WORLD_SIZE", world))))))
        node_id = int(env.get("GROUP_RANK", 0))
        return dict(rank=rank, local_rank=local, world_size=world,
                    n_nodes=nnodes, node_id=node_id,
                    id_run=env.get("TORCHELASTIC_RUN_ID", ""))
    if "SLURM_PROCID" in env:
        rank = int(env["SLURM_PROCID"])
        local = int(env.get("SLURM_LOCALID", 0))
        world = int(env.get("SLURM_NTASKS", 1))
        hostnames = expand_hostlist(env.get("SLURM_JOB_NODELIST", "localhost"))
        gpu_ids = env.get("SLURM_STEP_GPUS", "0").split(",")
        # master addr/port derivation mirrors reference trainer_base.py:147-153
        os.environ.setdefault("MASTER_ADDR", hostnames[0])
        os.environ.setdefault("MASTER_PORT", str(12346 + int(min(gpu_ids))))
        return dict(rank=rank, local_rank=local, world_size=world,
                    n_nodes=len(hostnames),
                    node_id=int(env.get("SLURM_NODEID", 0)),
                    id_run=env.get("SLURM_JOBID", ""))
    return dict(rank=0, local_rank=0, world_size=1, n_nodes=1, node_id=0,
                id_run="")


def init_distributed(backend: str | None = None,
                     timeout_s: int = 600) -> DistContext:
    # the host driver on this pool only supports dmabuf IPC; without this
    # RCCL cross-process CUDA-tensor sharing fails with
    # `hipIpcGetMemHandle: invalid argument`
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    topo = detect_topology()
    cuda = torch.cuda.is_available()
    if backend is None:
        backend = "nccl" if cuda else "gloo"
    if cuda:
        torch.cuda.set_device(topo["local_rank"] % max(1, torch.cuda.device_count()))
        device = torch.device("cuda", topo["local_rank"] % max(1, torch.cuda.device_count()))
    else:
        device = torch.device("cpu")

    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29517")
        dist.init_process_group(
            backend=backend,
            rank=topo["rank"],
            world_size=topo["world_size"],
            timeout=datetime.timedelta(seconds=timeout_s),
        )

    from acco_amd.utils.logging import create_id_run
    return DistContext(
        rank=topo["rank"], local_rank=topo["local_rank"],
        world_size=topo["world_size"], n_nodes=topo["n_nodes"],
        node_id=topo["node_id"], device=device, backend=backend,
        id_run=topo["id_run"] or create_id_run(),
    )
