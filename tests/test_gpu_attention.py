"""Flash attention (gfx950 MFMA) vs the fp32 torch reference — fwd + bwd,
causal, GQA, local window. Asymmetric random data (guide: symmetric inputs
mask transposed-layout bugs)."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def run_case(B, S, H, Hkv, D, window=None, scale=None, seed=0,
             atol=4e-2, rtol=4e-2):
    from acco_amd import ops
    from acco_amd.ops import torch_ref
    torch.manual_seed(seed)
    q = (torch.randn(B, S, H, D, device="cuda") * 0.5).bfloat16().requires_grad_(True)
    k = (torch.randn(B, S, Hkv, D, device="cuda") * 0.5).bfloat16().requires_grad_(True)
    v = (torch.randn(B, S, Hkv, D, device="cuda") * 0.5).bfloat16().requires_grad_(True)

    out = ops.causal_attention(q, k, v, scale=scale, window=window)
    assert out.shape == q.shape

    ref = torch_ref.causal_attention(q.detach().float(), k.detach().float(),
                                     v.detach().float(), scale=scale,
                                     window=window)
    err = (out.float() - ref).abs().max().item()
    assert torch.allclose(out.float(), ref, atol=atol, rtol=rtol), \
        f"fwd err {err}"

    dO = torch.randn_like(out)
    out.backward(dO)

    q32 = q.detach().float().requires_grad_(True)
    k32 = k.detach().float().requires_grad_(True)
    v32 = v.detach().float().requires_grad_(True)
    torch_ref.causal_attention(q32, k32, v32, scale=scale,
                               window=window).backward(dO.float())
    for name, got, want in [("dq", q.grad, q32.grad), ("dk", k.grad, k32.grad),
                            ("dv", v.grad, v32.grad)]:
        e = (got.float() - want).abs().max().item()
        assert torch.allclose(got.float(), want, atol=atol * 2, rtol=rtol), \
            f"{name} err {e}"


def test_attn_basic_d64():
    run_case(B=2, S=128, H=4, Hkv=4, D=64)


def test_attn_gqa_d64():
    run_case(B=2, S=192, H=8, Hkv=2, D=64, seed=1)


def test_attn_d128():
    run_case(B=1, S=128, H=4, Hkv=4, D=128, seed=2)


def test_attn_d128_v4_path():
    # S % 256 == 0 → the 32x32 v4 fwd/dq kernels with the v3 dkv
    run_case(B=1, S=256, H=4, Hkv=2, D=128, seed=7)


def test_attn_d64_v4_path_gqa_window():
    run_case(B=1, S=512, H=8, Hkv=2, D=64, window=96, seed=8)


def test_attn_local_window():
    # GPT-Neo-style: no scaling (1.0) + banded window; small magnitudes so
    # unscaled scores stay sane
    run_case(B=2, S=256, H=4, Hkv=4, D=64, window=64, scale=1.0, seed=3)


def test_attn_long_seq():
    run_case(B=1, S=1024, H=4, Hkv=2, D=64, seed=4)


def test_attn_used_by_llama_on_gpu():
    """The live model path must route through the flash kernel on GPU."""
    from acco_amd import ops
    assert ops.have_kernel("attn_fwd")


def test_packed_qkv_core_matches_standard_path():
    """The packed fused-QKV attention core (rope_packed + attn_*_packed,
    packed dqkv grad) against the standard per-tensor path, end to end
    through a small Llama block on GPU."""
    import torch
    from acco_amd.engine import arena
    from acco_amd.models import LlamaConfig, LlamaForCausalLM
    from acco_amd.models.fuse import install_fused_projections

    cfg = LlamaConfig(hidden_size=256, num_layers=2, num_heads=4,
                      num_kv_heads=2, intermediate_size=512, vocab_size=512,
                      max_position_embeddings=512)
    torch.manual_seed(0)
    m1 = LlamaForCausalLM(cfg)          # standard path
    torch.manual_seed(0)
    m2 = LlamaForCausalLM(cfg)          # fused + packed path

    dev = torch.device("cuda")
    p1 = arena.flatten_params(m1, torch.bfloat16, dev, pad_to=256)
    g1 = arena.attach_grad_arena(m1, torch.bfloat16, dev, pad_to=256)
    p2 = arena.flatten_params(m2, torch.bfloat16, dev, pad_to=256)
    g2 = arena.attach_grad_arena(m2, torch.bfloat16, dev, pad_to=256)
    assert torch.equal(p1, p2)
    n_f = install_fused_projections(m2, p2, g2)
    assert n_f > 0

    ids = torch.randint(0, 512, (2, 256), device=dev)
    loss1, _ = m1(ids, labels=ids)
    loss1.backward()
    loss2, _ = m2(ids, labels=ids)
    loss2.backward()
    assert abs(float(loss1) - float(loss2)) < 2e-2, (loss1, loss2)
    n = arena.live_numel(m1)
    cos = torch.nn.functional.cosine_similarity(g1[:n].float(),
                                                g2[:n].float(), dim=0)
    assert cos > 0.995, f"grad cosine {cos}"


def test_packed_kvq_core_gptneo_matches_standard():
    """GPT-Neo packed path (k|v|q order, no rope, scale=1, local window)
    vs the standard path, end to end on GPU."""
    import torch
    from acco_amd.engine import arena
    from acco_amd.models import GPTNeoConfig, GPTNeoForCausalLM
    from acco_amd.models.fuse import install_fused_projections

    cfg = GPTNeoConfig(hidden_size=256, num_layers=2, num_heads=4,
                       vocab_size=512, max_position_embeddings=512,
                       window_size=64)
    torch.manual_seed(0)
    m1 = GPTNeoForCausalLM(cfg)
    torch.manual_seed(0)
    m2 = GPTNeoForCausalLM(cfg)
    dev = torch.device("cuda")
    p1 = arena.flatten_params(m1, torch.bfloat16, dev, pad_to=256)
    g1 = arena.attach_grad_arena(m1, torch.bfloat16, dev, pad_to=256)
    p2 = arena.flatten_params(m2, torch.bfloat16, dev, pad_to=256)
    g2 = arena.attach_grad_arena(m2, torch.bfloat16, dev, pad_to=256)
    assert install_fused_projections(m2, p2, g2) > 0

    ids = torch.randint(0, 512, (2, 256), device=dev)
    loss1, _ = m1(ids, labels=ids)
    loss1.backward()
    loss2, _ = m2(ids, labels=ids)
    loss2.backward()
    assert abs(float(loss1) - float(loss2)) < 2e-2, (loss1, loss2)
    n = arena.live_numel(m1)
    cos = torch.nn.functional.cosine_similarity(g1[:n].float(),
                                                g2[:n].float(), dim=0)
    assert cos > 0.995, f"grad cosine {cos}"


@pytest.mark.gpu
def test_packed_qkv_core_llama_d128_matches_standard():
    """Packed attention core at head_dim=128 (llama-8b geometry): fused
    projection + packed rope/attention/grad vs the standard path."""
    import torch
    from acco_amd.engine import arena
    from acco_amd.models import LlamaConfig, LlamaForCausalLM
    from acco_amd.models.fuse import install_fused_projections

    cfg = LlamaConfig(hidden_size=256, num_layers=2, num_heads=2,
                      num_kv_heads=1, intermediate_size=512, vocab_size=512,
                      max_position_embeddings=512)
    assert cfg.head_dim == 128
    torch.manual_seed(0)
    m1 = LlamaForCausalLM(cfg)
    torch.manual_seed(0)
    m2 = LlamaForCausalLM(cfg)

    dev = torch.device("cuda")
    p1 = arena.flatten_params(m1, torch.bfloat16, dev, pad_to=256)
    g1 = arena.attach_grad_arena(m1, torch.bfloat16, dev, pad_to=256)
    p2 = arena.flatten_params(m2, torch.bfloat16, dev, pad_to=256)
    g2 = arena.attach_grad_arena(m2, torch.bfloat16, dev, pad_to=256)
    assert torch.equal(p1, p2)
    assert install_fused_projections(m2, p2, g2) > 0

    ids = torch.randint(0, 512, (2, 256), device=dev)
    loss1, _ = m1(ids, labels=ids)
    loss1.backward()
    loss2, _ = m2(ids, labels=ids)
    loss2.backward()
    assert abs(float(loss1) - float(loss2)) < 2e-2, (loss1, loss2)
    n = arena.live_numel(m1)
    cos = torch.nn.functional.cosine_similarity(g1[:n].float(),
                                                g2[:n].float(), dim=0)
    assert cos > 0.995, f"grad cosine {cos}"


def test_attn_defer_max_rescale_forced():
    """T13 hazard test (guide §5.4 rule 26): the defer-max branch is rare on
    random data — force a LATE rescale by spiking K rows deep into the
    sequence so the running max jumps past the threshold at chosen tiles,
    after O has accumulated. Checked against the fp32 reference."""
    from acco_amd import ops
    from acco_amd.ops import torch_ref
    torch.manual_seed(7)
    B, S, H, Hkv, D = 2, 1024, 8, 4, 64
    q = (torch.randn(B, S, H, D, device="cuda") * 0.5).bfloat16()
    k = (torch.randn(B, S, Hkv, D, device="cuda") * 0.5).bfloat16()
    v = (torch.randn(B, S, Hkv, D, device="cuda") * 0.5).bfloat16()
    # spike several K rows at different tiles: q·k for these rows ≫ any
    # other score (|q·k| ≲ scale·D·0.25 ≈ 2 normally; the spike gives
    # ≈ 0.125·8·64·0.5·~q ≫ 8/ln2 growth), so every q row ≥ the spike
    # position must rescale its accumulated O there
    for pos in (300, 650, 900):
        k[:, pos, :, :] = 8.0
    q.requires_grad_(True)
    k.requires_grad_(True)
    v.requires_grad_(True)

    out = ops.causal_attention(q, k, v)
    ref = torch_ref.causal_attention(q.detach().float(), k.detach().float(),
                                     v.detach().float())
    assert torch.allclose(out.float(), ref, atol=5e-2, rtol=5e-2), \
        (out.float() - ref).abs().max()

    dO = torch.randn_like(out)
    out.backward(dO)
    q32 = q.detach().float().requires_grad_(True)
    k32 = k.detach().float().requires_grad_(True)
    v32 = v.detach().float().requires_grad_(True)
    torch_ref.causal_attention(q32, k32, v32).backward(dO.float())
    for name, got, want in [("dq", q.grad, q32.grad),
                            ("dk", k.grad, k32.grad),
                            ("dv", v.grad, v32.grad)]:
        e = (got.float() - want).abs().max().item()
        assert torch.allclose(got.float(), want, atol=8e-2, rtol=8e-2), \
            f"{name} err {e}"
