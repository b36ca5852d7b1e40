"""Fused arena-view QKV / gate-up projections: same forward and same
accumulated gradients as the unfused per-projection path."""

import torch

from acco_amd.engine import arena
from acco_amd.models import (GPTNeoConfig, GPTNeoForCausalLM, LlamaConfig,
                             LlamaForCausalLM)
from acco_amd.models.fuse import install_fused_projections


def _grad_check(model_fn):
    torch.manual_seed(0)
    m1 = model_fn()
    torch.manual_seed(0)
    m2 = model_fn()
    dev = torch.device("cpu")
    n = arena.live_numel(m1)

    p1 = arena.flatten_params(m1, torch.float32, dev, pad_to=256)
    g1 = arena.attach_grad_arena(m1, torch.float32, dev, pad_to=256)
    p2 = arena.flatten_params(m2, torch.float32, dev, pad_to=256)
    g2 = arena.attach_grad_arena(m2, torch.float32, dev, pad_to=256)
    assert torch.equal(p1, p2)

    n_fused = install_fused_projections(m2, p2, g2)
    assert n_fused > 0

    ids = torch.randint(0, 64, (2, 16))
    for _ in range(2):                     # accumulate two backwards
        loss1, _ = m1(ids, labels=ids)
        loss1.backward()
        loss2, _ = m2(ids, labels=ids)
        loss2.backward()
        assert torch.allclose(loss1, loss2, atol=1e-6)
    assert torch.allclose(g1[:n], g2[:n], atol=1e-5, rtol=1e-5), \
        (g1 - g2).abs().max()


def test_llama_fused_grads_match():
    cfg = LlamaConfig(hidden_size=32, num_layers=2, num_heads=4,
                      num_kv_heads=2, intermediate_size=64, vocab_size=64,
                      max_position_embeddings=64)
    _grad_check(lambda: LlamaForCausalLM(cfg))


def test_gptneo_fused_grads_match():
    cfg = GPTNeoConfig(hidden_size=32, num_layers=2, num_heads=2,
                       vocab_size=64, max_position_embeddings=32,
                       window_size=8)
    _grad_check(lambda: GPTNeoForCausalLM(cfg))
