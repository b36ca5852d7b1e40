"""Per-layer activation recompute: identical loss and identical accumulated
gradients (the recompute path re-runs the in-place-dW arena Functions —
they must accumulate exactly once per backward)."""

import torch

from acco_amd.engine import arena
from acco_amd.models import (GPTNeoConfig, GPTNeoForCausalLM, LlamaConfig,
                             LlamaForCausalLM)
from acco_amd.models.fuse import install_fused_projections


def _check(model_fn, inner_attr):
    torch.manual_seed(3)
    m1 = model_fn()
    torch.manual_seed(3)
    m2 = model_fn()
    dev = torch.device("cpu")
    p1 = arena.flatten_params(m1, torch.float32, dev, pad_to=256)
    g1 = arena.attach_grad_arena(m1, torch.float32, dev, pad_to=256)
    p2 = arena.flatten_params(m2, torch.float32, dev, pad_to=256)
    g2 = arena.attach_grad_arena(m2, torch.float32, dev, pad_to=256)
    install_fused_projections(m1, p1, g1)
    install_fused_projections(m2, p2, g2)
    getattr(m2, inner_attr).gradient_checkpointing = True

    ids = torch.randint(0, 64, (2, 16))
    for _ in range(2):                     # two accumulating backwards
        l1, _ = m1(ids, labels=ids)
        l1.backward()
        l2, _ = m2(ids, labels=ids)
        l2.backward()
        assert torch.allclose(l1, l2, atol=1e-6)
    n = arena.live_numel(m1)
    assert torch.allclose(g1[:n], g2[:n], atol=1e-5, rtol=1e-5), \
        (g1 - g2).abs().max()


def test_llama_checkpointing_grads_match():
    cfg = LlamaConfig(hidden_size=32, num_layers=2, num_heads=4,
                      num_kv_heads=2, intermediate_size=64, vocab_size=64,
                      max_position_embeddings=64)
    _check(lambda: LlamaForCausalLM(cfg), "model")


def test_gptneo_checkpointing_grads_match():
    cfg = GPTNeoConfig(hidden_size=32, num_layers=2, num_heads=2,
                       vocab_size=64, max_position_embeddings=32,
                       window_size=8)
    _check(lambda: GPTNeoForCausalLM(cfg), "transformer")
