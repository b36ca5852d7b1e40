"""Multi-node-SHAPED bootstrap: two processes faking two SLURM nodes
(SLURM_PROCID/NODEID/NTASKS + a 2-host nodelist) go end-to-end through
init_distributed → CommBackend → an ACCO training round. Covers the SLURM
branch of detect_topology the way a real 2×1-GPU sbatch would drive it
(reference slurm2.slurm:1-31), with the master on 127.0.0.1 so rendezvous
works inside one container."""

import os

import torch
import torch.nn as nn

from tests.conftest import run_distributed
from tests.dist_utils import teardown_worker


def _worker(rank, world, port, tmpdir):
    # wipe torchrun-style vars so the SLURM branch is the one exercised
    for k in ("RANK", "WORLD_SIZE", "LOCAL_RANK", "MASTER_ADDR",
              "MASTER_PORT"):
        os.environ.pop(k, None)
    os.environ["SLURM_PROCID"] = str(rank)
    os.environ["SLURM_LOCALID"] = "0"           # 1 task per fake node
    os.environ["SLURM_NTASKS"] = str(world)
    os.environ["SLURM_NODEID"] = str(rank)
    os.environ["SLURM_JOB_NODELIST"] = "127.0.0.[1-2]"
    os.environ["SLURM_JOBID"] = "424242"
    # port derivation mirrors trainer_base.py:153 (12346 + min gpu id);
    # feed it the test's free port so parallel runs don't collide
    os.environ["SLURM_STEP_GPUS"] = str(port - 12346)
    # nodelist host "127.0.0.1" is the master — reachable in-container

    from acco_amd.engine.bootstrap import init_distributed
    ctx = init_distributed()
    assert ctx.rank == rank
    assert ctx.world_size == world
    assert ctx.local_rank == 0
    assert ctx.n_nodes == 2
    assert ctx.node_id == rank
    assert ctx.id_run == "424242"
    assert os.environ["MASTER_ADDR"] == "127.0.0.1"
    assert os.environ["MASTER_PORT"] == str(port)

    # end-to-end: a tiny ACCO round over the bootstrapped communicator
    from acco_amd.engine import arena
    from acco_amd.engine.acco import AccoEngine
    from acco_amd.engine.scheduler import LRSchedule
    from acco_amd.engine.sharded_adamw import ShardedAdamW
    from acco_amd.parallel.comm import CommBackend, ShardSpec

    torch.manual_seed(5)
    model = nn.Sequential(nn.Linear(8, 8), nn.Linear(8, 4))
    n = arena.live_numel(model)
    spec = ShardSpec.build(n, world, buckets=2, align=4)
    params = arena.flatten_params(model, torch.float32, ctx.device,
                                  pad_to=spec.total)
    grads = arena.attach_grad_arena(model, torch.float32, ctx.device,
                                    pad_to=spec.total)
    comm = CommBackend(ctx.device)
    comm.all_reduce_avg(params)
    opt = ShardedAdamW(spec, rank, ctx.device, lr=1e-2)
    opt.init_master_from_buffer(params)

    def fb(inputs):
        x, y = inputs
        loss = ((model(x) - y) ** 2).mean()
        loss.backward()
        return loss.detach()

    gen = torch.Generator().manual_seed(100 + rank)

    def nb():
        return (torch.randn(4, 8, generator=gen),
                torch.randn(4, 4, generator=gen))

    eng = AccoEngine(params_arena=params, grads_arena=grads, n_live=n,
                     spec=spec, comm=comm, rank=rank, device=ctx.device,
                     opt=opt, sched=LRSchedule(1e-2, 0, 100, "constant"),
                     forward_backward=fb, next_batch=nb,
                     n_grad_accumulation=1)
    eng.train_acco(nb_grad_tot=1 << 30, max_rounds=4)
    torch.save(params[:n].clone(), os.path.join(tmpdir, f"p_{rank}.pt"))
    teardown_worker()


def test_two_fake_slurm_nodes_end_to_end():
    world = 2
    tmpdir = run_distributed(_worker, world, timeout=240)
    p0 = torch.load(os.path.join(tmpdir, "p_0.pt"), weights_only=False)
    p1 = torch.load(os.path.join(tmpdir, "p_1.pt"), weights_only=False)
    assert torch.equal(p0, p1), "ranks diverged after the all-gather"
    assert torch.isfinite(p0).all()
