"""End-to-end GPU integration: the full model (all HIP kernels composed)
against the fp32 CPU reference — loss and a gradient-direction check."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def _run(model_fn, ids):
    from acco_amd.engine import arena
    torch.manual_seed(0)
    m_cpu = model_fn()
    torch.manual_seed(0)
    m_gpu = model_fn()

    loss_c, _ = m_cpu(ids, labels=ids)
    loss_c.backward()
    g_cpu = torch.cat([p.grad.reshape(-1) for p in m_cpu.parameters()])

    m_gpu = m_gpu.to("cuda", dtype=torch.bfloat16)
    ids_g = ids.cuda()
    loss_g, _ = m_gpu(ids_g, labels=ids_g)
    loss_g.backward()
    g_gpu = torch.cat([p.grad.reshape(-1).float() for p in m_gpu.parameters()]).cpu()
    del arena
    return float(loss_c), float(loss_g), g_cpu, g_gpu


def _check(loss_c, loss_g, g_cpu, g_gpu):
    assert abs(loss_c - loss_g) / max(abs(loss_c), 1e-6) < 0.05, \
        (loss_c, loss_g)
    cos = torch.nn.functional.cosine_similarity(g_cpu, g_gpu, dim=0)
    assert cos > 0.98, f"grad cosine {cos}"


def test_llama_full_model_gpu_vs_cpu():
    from acco_amd.models import LlamaConfig, LlamaForCausalLM
    cfg = LlamaConfig(hidden_size=256, num_layers=2, num_heads=4,
                      num_kv_heads=2, intermediate_size=512, vocab_size=512,
                      max_position_embeddings=256)
    ids = torch.randint(0, 512, (2, 128))
    _check(*_run(lambda: LlamaForCausalLM(cfg), ids))


def test_gptneo_full_model_gpu_vs_cpu():
    from acco_amd.models import GPTNeoConfig, GPTNeoForCausalLM
    cfg = GPTNeoConfig(hidden_size=128, num_layers=2, num_heads=2,
                       vocab_size=512, max_position_embeddings=256,
                       window_size=32)
    ids = torch.randint(0, 512, (2, 128))
    _check(*_run(lambda: GPTNeoForCausalLM(cfg), ids))
