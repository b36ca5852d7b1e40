"""NativeZeroDDP (bucketed reduce-scatter + ZeRO-1 fused AdamW + all-gather)
vs a plain single-process AdamW oracle on averaged gradients."""

import os

import torch
import torch.nn as nn

from tests.conftest import run_distributed
from tests.dist_utils import init_worker, teardown_worker

D_IN, D_OUT = 6, 3
STEPS = 5
N_ACC = 2
LR = 1e-2


def make_model():
    torch.manual_seed(11)
    return nn.Linear(D_IN, D_OUT, bias=True)


def make_batch(rank: int, step: int, micro: int):
    g = torch.Generator().manual_seed(rank * 10007 + step * 101 + micro)
    x = torch.randn(5, D_IN, generator=g)
    y = torch.randn(5, D_OUT, generator=g)
    return x, y


def loss_fn(model, batch):
    x, y = batch
    return ((model(x) - y) ** 2).mean()


def _worker_ddp(rank, world, port, tmpdir):
    init_worker(rank, world, port)
    from acco_amd.engine import arena
    from acco_amd.engine.sharded_adamw import ShardedAdamW
    from acco_amd.parallel.comm import CommBackend, ShardSpec
    from acco_amd.parallel.ddp import NativeZeroDDP

    model = make_model()
    device = torch.device("cpu")
    n = arena.live_numel(model)
    spec = ShardSpec.build(n, world, buckets=2, align=2)
    params = arena.flatten_params(model, torch.float32, device, pad_to=spec.total)
    grads = arena.attach_grad_arena(model, torch.float32, device, pad_to=spec.total)
    comm = CommBackend(device)
    opt = ShardedAdamW(spec, rank, device, lr=LR, weight_decay=0.01)
    opt.init_master_from_buffer(params)
    ddp = NativeZeroDDP(model, params, grads, n, spec, comm, rank, opt)

    for step in range(STEPS):
        for micro in range(N_ACC):
            if micro == N_ACC - 1:
                ddp.begin_sync_microbatch()
            loss = loss_fn(model, make_batch(rank, step, micro)) / N_ACC
            loss.backward()
        ddp.finish_step(grad_scale=1.0 / world)
        ddp.zero_grad()

    torch.save(params[:n].clone(), os.path.join(tmpdir, f"p_{rank}.pt"))
    teardown_worker()


def _worker_ddp_fused(rank, world, port, tmpdir, fused):
    """Tiny tied-weight Llama through NativeZeroDDP, fused vs unfused
    projections — the element-coverage readiness path (parallel/ddp.py)
    must launch every bucket from inside backward and produce the same
    update as the plain per-param-hook path."""
    init_worker(rank, world, port)
    from acco_amd.engine import arena
    from acco_amd.engine.sharded_adamw import ShardedAdamW
    from acco_amd.models import LlamaConfig, LlamaForCausalLM
    from acco_amd.models.fuse import install_fused_projections
    from acco_amd.parallel.comm import CommBackend, ShardSpec
    from acco_amd.parallel.ddp import NativeZeroDDP

    cfg = LlamaConfig(hidden_size=32, num_layers=2, num_heads=4,
                      num_kv_heads=2, intermediate_size=64, vocab_size=64,
                      max_position_embeddings=64, tie_word_embeddings=True)
    torch.manual_seed(7)
    model = LlamaForCausalLM(cfg)
    device = torch.device("cpu")
    n = arena.live_numel(model)
    spec = ShardSpec.build(n, world, buckets=4, align=8)
    params = arena.flatten_params(model, torch.float32, device, pad_to=spec.total)
    grads = arena.attach_grad_arena(model, torch.float32, device, pad_to=spec.total)
    comm = CommBackend(device)
    if fused:
        assert install_fused_projections(model, params, grads) > 0
    opt = ShardedAdamW(spec, rank, device, lr=LR, weight_decay=0.01)
    opt.init_master_from_buffer(params)
    ddp = NativeZeroDDP(model, params, grads, n, spec, comm, rank, opt)

    for step in range(3):
        for micro in range(N_ACC):
            if micro == N_ACC - 1:
                ddp.begin_sync_microbatch()
            g = torch.Generator().manual_seed(rank * 131 + step * 17 + micro)
            ids = torch.randint(0, cfg.vocab_size, (2, 16), generator=g)
            loss, _ = model(ids, labels=ids)
            (loss / N_ACC).backward()
        # coverage accounting must have launched every live bucket from
        # inside backward (overlap), not left them to finish_step
        for b in range(spec.nb):
            if ddp._req[b] > 0:
                assert ddp._launched[b], f"bucket {b} not launched in backward"
        ddp.finish_step(grad_scale=1.0 / world)
        ddp.zero_grad()

    torch.save(params[:n].clone(),
               os.path.join(tmpdir, f"p_{'f' if fused else 'u'}_{rank}.pt"))
    teardown_worker()


def test_native_ddp_fused_projections_match_unfused_ws2():
    world = 2
    d1 = run_distributed(_worker_ddp_fused, world, args=(True,), timeout=240)
    d2 = run_distributed(_worker_ddp_fused, world, args=(False,), timeout=240)
    pf = torch.load(os.path.join(d1, "p_f_0.pt"), weights_only=False)
    pf1 = torch.load(os.path.join(d1, "p_f_1.pt"), weights_only=False)
    pu = torch.load(os.path.join(d2, "p_u_0.pt"), weights_only=False)
    assert torch.equal(pf, pf1)          # ranks agree after all-gather
    assert torch.allclose(pf, pu, atol=1e-5, rtol=1e-5), \
        (pf - pu).abs().max()


def test_native_ddp_matches_adamw_oracle_ws2():
    world = 2
    tmpdir = run_distributed(_worker_ddp, world, timeout=180)
    res = [torch.load(os.path.join(tmpdir, f"p_{r}.pt"), weights_only=False)
           for r in range(world)]
    assert torch.equal(res[0], res[1])

    # oracle: single-process AdamW on rank-averaged accumulated grads
    model = make_model()
    opt = torch.optim.AdamW(model.parameters(), lr=LR, betas=(0.9, 0.95),
                            eps=1e-8, weight_decay=0.01)
    for step in range(STEPS):
        opt.zero_grad()
        for rank in range(world):
            for micro in range(N_ACC):
                loss = loss_fn(model, make_batch(rank, step, micro))
                (loss / (N_ACC * world)).backward()
        opt.step()
    flat = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    assert torch.allclose(res[0], flat, atol=1e-6, rtol=1e-5), \
        (res[0] - flat).abs().max()


def test_native_ddp_fused_ws3_ragged():
    """world=3 (shards don't divide evenly): coverage-based bucket launch +
    ragged sharded AdamW + all-gather keep all three ranks bitwise equal."""
    d = run_distributed(_worker_ddp_fused, 3, args=(True,), timeout=240)
    ps = [torch.load(os.path.join(d, f"p_f_{r}.pt"), weights_only=False)
          for r in range(3)]
    assert torch.equal(ps[0], ps[1]) and torch.equal(ps[0], ps[2])
    assert torch.isfinite(ps[0]).all()
