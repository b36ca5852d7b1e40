"""Arena-swap (zero-copy params⇄buffer handover) must be bit-identical to
the copy path, with fused projection views flipping alongside."""

import torch
import torch.nn as nn

from acco_amd.engine import arena
from acco_amd.engine.acco import AccoEngine
from acco_amd.engine.scheduler import LRSchedule
from acco_amd.engine.sharded_adamw import ShardedAdamW
from acco_amd.models import LlamaConfig, LlamaForCausalLM
from acco_amd.models.fuse import install_fused_projections
from acco_amd.parallel.comm import CommBackend, ShardSpec


def _run(swap: bool, method: str):
    cfg = LlamaConfig(hidden_size=32, num_layers=2, num_heads=4,
                      num_kv_heads=2, intermediate_size=64, vocab_size=64,
                      max_position_embeddings=64)
    torch.manual_seed(21)
    model = LlamaForCausalLM(cfg)
    dev = torch.device("cpu")
    n = arena.live_numel(model)
    spec = ShardSpec.build(n, 1, buckets=2, align=8)
    params = arena.flatten_params(model, torch.float32, dev, pad_to=spec.total)
    grads = arena.attach_grad_arena(model, torch.float32, dev,
                                    pad_to=spec.total)
    install_fused_projections(model, params, grads)
    comm = CommBackend(dev)
    opt = ShardedAdamW(spec, 0, dev, lr=1e-2)
    opt.init_master_from_buffer(params)
    gen = torch.Generator().manual_seed(33)

    def nb():
        return {"ids": torch.randint(0, cfg.vocab_size, (2, 16),
                                     generator=gen)}

    def fb(inputs):
        loss, _ = model(inputs["ids"], labels=inputs["ids"])
        loss.backward()
        return loss.detach()

    eng = AccoEngine(params_arena=params, grads_arena=grads, n_live=n,
                     spec=spec, comm=comm, rank=0, device=dev, opt=opt,
                     sched=LRSchedule(1e-2, 0, 100, "constant"),
                     forward_backward=fb, next_batch=nb,
                     n_grad_accumulation=1)
    if swap:
        eng.enable_arena_swap(model, grads)
    if method == "acco":
        eng.train_acco(nb_grad_tot=1 << 30, max_rounds=5)
    else:
        eng.train_dpu(nb_grad_tot=1 << 30, max_rounds=5)
    return eng.params[:n].clone(), eng


def test_arena_swap_acco_sane():
    """ACCO's comm thread makes the per-round grad schedule timing-
    dependent, so swap-vs-copy can't be compared bitwise here (the DPU
    test below is the deterministic bitwise vehicle; the ws2 trainer
    integration tests and the oracle cover ACCO itself). This checks the
    swap machinery engaged and left a consistent model."""
    p_swap, eng = _run(True, "acco")
    assert eng.flip_param_views is not None
    assert torch.isfinite(p_swap).all()


def test_arena_swap_bitwise_equals_copy_path_dpu():
    """DPU is sequential/deterministic: the role-swap path must be
    bit-identical to the copy path."""
    p_copy, _ = _run(False, "dpu")
    p_swap, eng = _run(True, "dpu")
    assert eng.flip_param_views is not None
    assert torch.equal(p_copy, p_swap)


def test_repoint_params_preserves_values_and_order():
    model = nn.Sequential(nn.Linear(4, 4), nn.Linear(4, 2))
    dev = torch.device("cpu")
    n = arena.live_numel(model)
    a = arena.flatten_params(model, torch.float32, dev, pad_to=8)
    b = torch.full_like(a, 7.0)
    before = [p.clone() for p in model.parameters()]
    arena.repoint_params(model, b)
    assert all(torch.all(p == 7.0) for p in model.parameters())
    arena.repoint_params(model, a)
    for p, want in zip(model.parameters(), before):
        assert torch.equal(p, want)
