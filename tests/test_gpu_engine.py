"""GPU (MI355X) end-to-end checks: extension loads, models step in bf16,
the ACCO engine runs single-GPU with the fused HIP AdamW."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_hip_ext_loaded():
    from acco_amd import ops
    assert ops.ext_available(), "in-tree _hip_ops.so must be present on GPU"
    assert ops.have_kernel("fused_adamw")


def test_llama_forward_backward_bf16():
    from acco_amd.models import LlamaConfig, LlamaForCausalLM
    torch.manual_seed(0)
    cfg = LlamaConfig(hidden_size=256, num_layers=2, num_heads=4,
                      num_kv_heads=2, intermediate_size=512, vocab_size=1024,
                      max_position_embeddings=512)
    model = LlamaForCausalLM(cfg).to("cuda", dtype=torch.bfloat16)
    ids = torch.randint(0, 1024, (2, 128), device="cuda")
    loss, logits = model(ids, labels=ids)
    loss.backward()
    torch.cuda.synchronize()
    assert torch.isfinite(loss)
    assert torch.isfinite(logits.float()).all()


def test_gptneo_forward_backward_bf16():
    from acco_amd.models import GPTNeoConfig, GPTNeoForCausalLM
    torch.manual_seed(0)
    cfg = GPTNeoConfig(hidden_size=128, num_layers=2, num_heads=4,
                       vocab_size=512, max_position_embeddings=256,
                       window_size=32)
    model = GPTNeoForCausalLM(cfg).to("cuda", dtype=torch.bfloat16)
    ids = torch.randint(0, 512, (2, 64), device="cuda")
    loss, _ = model(ids, labels=ids)
    loss.backward()
    torch.cuda.synchronize()
    assert torch.isfinite(loss)


def test_acco_engine_single_gpu():
    from acco_amd.engine import arena
    from acco_amd.engine.acco import AccoEngine
    from acco_amd.engine.scheduler import LRSchedule
    from acco_amd.engine.sharded_adamw import ShardedAdamW
    from acco_amd.models import LlamaConfig, LlamaForCausalLM
    from acco_amd.parallel.comm import CommBackend, ShardSpec

    torch.manual_seed(0)
    device = torch.device("cuda")
    cfg = LlamaConfig(hidden_size=256, num_layers=2, num_heads=4,
                      num_kv_heads=2, intermediate_size=512, vocab_size=1024,
                      max_position_embeddings=512)
    model = LlamaForCausalLM(cfg)
    n = arena.live_numel(model)
    spec = ShardSpec.build(n, 1, buckets=4)
    params = arena.flatten_params(model, torch.bfloat16, device,
                                  pad_to=spec.total)
    grads = arena.attach_grad_arena(model, torch.bfloat16, device,
                                    pad_to=spec.total)
    comm = CommBackend(device)
    opt = ShardedAdamW(spec, 0, device, lr=1e-3)
    opt.init_master_from_buffer(params)
    sched = LRSchedule(1e-3, 10, 1000, "cosine")

    ids = torch.randint(0, 1024, (2, 128), device=device)

    def forward_backward(_):
        loss, _l = model(ids, labels=ids)
        loss.backward()
        return loss.detach()

    eng = AccoEngine(params_arena=params, grads_arena=grads, n_live=n,
                     spec=spec, comm=comm, rank=0, device=device, opt=opt,
                     sched=sched, forward_backward=forward_backward,
                     next_batch=lambda: {}, n_grad_accumulation=1)
    p0 = params[:n].clone()
    eng.train_acco(nb_grad_tot=1 << 60, max_rounds=6)
    torch.cuda.synchronize()
    assert eng.round_idx == 6
    assert opt.step_count == 3          # every odd round commits
    assert torch.isfinite(params[:n].float()).all()
    assert not torch.equal(params[:n], p0)
    assert torch.isfinite(eng.loss_static).all()
