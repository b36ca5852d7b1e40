"""Topology detection (reference initialize_com trainer_base.py:135-153
capability): torchrun env, SLURM env (with hostlist + master derivation),
single-process fallback."""

import os
from unittest import mock

from acco_amd.engine.bootstrap import detect_topology


def _clean_env(extra):
    keys = ["RANK", "LOCAL_RANK", "WORLD_SIZE", "NNODES", "GROUP_RANK",
            "LOCAL_WORLD_SIZE", "SLURM_PROCID", "SLURM_LOCALID",
            "SLURM_NTASKS", "SLURM_JOB_NODELIST", "SLURM_STEP_GPUS",
            "SLURM_NODEID", "SLURM_JOBID", "MASTER_ADDR", "MASTER_PORT",
            "TORCHELASTIC_RUN_ID"]
    env = {k: v for k, v in os.environ.items() if k not in keys}
    env.update(extra)
    return env


def test_torchrun_env():
    env = _clean_env({"RANK": "3", "LOCAL_RANK": "3", "WORLD_SIZE": "8",
                      "LOCAL_WORLD_SIZE": "8"})
    with mock.patch.dict(os.environ, env, clear=True):
        t = detect_topology()
    assert t["rank"] == 3 and t["local_rank"] == 3 and t["world_size"] == 8


def test_slurm_env_master_derivation():
    env = _clean_env({"SLURM_PROCID": "9", "SLURM_LOCALID": "1",
                      "SLURM_NTASKS": "16",
                      "SLURM_JOB_NODELIST": "gpu[01-02]",
                      "SLURM_STEP_GPUS": "2,3", "SLURM_NODEID": "1",
                      "SLURM_JOBID": "4242"})
    with mock.patch.dict(os.environ, env, clear=True):
        t = detect_topology()
        # master addr/port derived as in reference trainer_base.py:147-153
        assert os.environ["MASTER_ADDR"] == "gpu01"
        assert os.environ["MASTER_PORT"] == str(12346 + 2)
    assert t["rank"] == 9 and t["world_size"] == 16
    assert t["n_nodes"] == 2 and t["node_id"] == 1
    assert t["id_run"] == "4242"


def test_single_process_fallback():
    with mock.patch.dict(os.environ, _clean_env({}), clear=True):
        t = detect_topology()
    assert t == dict(rank=0, local_rank=0, world_size=1, n_nodes=1,
                     node_id=0, id_run="")
