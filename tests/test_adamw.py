"""Numerics of the fused AdamW (torch_ref path) vs torch.optim.AdamW, and
the tentative (commit=False) mode vs snapshot→step→rollback."""

import torch
import pytest

from acco_amd.ops import torch_ref


def run_ref_steps(p0, grads, lr=1e-3, b1=0.9, b2=0.95, eps=1e-8, wd=0.1):
    p = p0.clone()
    m = torch.zeros_like(p)
    v = torch.zeros_like(p)
    outs = []
    for t, g in enumerate(grads):
        out = torch.empty_like(p, dtype=torch.bfloat16)
        torch_ref.fused_adamw_step(p, g, m, v, t, lr, b1, b2, eps, wd,
                                   grad_scale=1.0, out_bf16=out, commit=True)
        outs.append(out)
    return p, outs


def test_matches_torch_adamw():
    torch.manual_seed(0)
    n = 257
    p0 = torch.randn(n)
    grads = [torch.randn(n) for _ in range(5)]

    p_ref, _ = run_ref_steps(p0, grads)

    p_t = p0.clone().requires_grad_(True)
    opt = torch.optim.AdamW([p_t], lr=1e-3, betas=(0.9, 0.95), eps=1e-8,
                            weight_decay=0.1)
    for g in grads:
        p_t.grad = g.clone()
        opt.step()
    assert torch.allclose(p_ref, p_t.detach(), atol=1e-6, rtol=1e-5)


def test_grad_scale_tensor_and_cast():
    torch.manual_seed(1)
    n = 64
    p0 = torch.randn(n)
    g_bf16 = torch.randn(n).bfloat16()
    scale_t = torch.tensor([0.25])

    p1 = p0.clone(); m1 = torch.zeros(n); v1 = torch.zeros(n)
    torch_ref.fused_adamw_step(p1, g_bf16, m1, v1, 0, 1e-3, 0.9, 0.95, 1e-8,
                               0.0, grad_scale=scale_t, commit=True)

    p2 = p0.clone(); m2 = torch.zeros(n); v2 = torch.zeros(n)
    torch_ref.fused_adamw_step(p2, g_bf16.float() * 0.25, m2, v2, 0, 1e-3,
                               0.9, 0.95, 1e-8, 0.0, grad_scale=1.0,
                               commit=True)
    assert torch.allclose(p1, p2)


def test_tentative_equals_snapshot_step_rollback():
    """commit=False must equal the reference's clone→step→restore
    (trainer_decoupled.py:79-84,113-125) while leaving state untouched."""
    torch.manual_seed(2)
    n = 128
    p = torch.randn(n)
    m = torch.randn(n).abs() * 0.01
    v = torch.randn(n).abs() * 0.001
    g = torch.randn(n)
    step = 3

    p_snap, m_snap, v_snap = p.clone(), m.clone(), v.clone()

    # snapshot → commit step → capture out → rollback (reference way)
    out_ref = torch.empty(n, dtype=torch.bfloat16)
    p2, m2, v2 = p.clone(), m.clone(), v.clone()
    torch_ref.fused_adamw_step(p2, g, m2, v2, step, 1e-3, 0.9, 0.95, 1e-8,
                               0.1, grad_scale=0.5, out_bf16=out_ref,
                               commit=True)

    # tentative mode (ours)
    out = torch.empty(n, dtype=torch.bfloat16)
    torch_ref.fused_adamw_step(p, g, m, v, step, 1e-3, 0.9, 0.95, 1e-8, 0.1,
                               grad_scale=0.5, out_bf16=out, commit=False)

    assert torch.equal(out, out_ref)
    assert torch.equal(p, p_snap) and torch.equal(m, m_snap) and torch.equal(v, v_snap)


@pytest.mark.gpu
def test_hip_fused_adamw_matches_ref():
    """HIP kernel vs fp32 torch reference on GPU."""
    from acco_amd import ops
    torch.manual_seed(3)
    n = 4096 + 256
    dev = "cuda"
    p = torch.randn(n, device=dev)
    m = torch.rand(n, device=dev) * 0.01
    v = torch.rand(n, device=dev) * 0.001
    g = torch.randn(n, device=dev).bfloat16()
    scale_t = torch.tensor([1.0 / 3.0], device=dev)

    p_ref, m_ref, v_ref = p.clone(), m.clone(), v.clone()
    out_ref = torch.empty(n, device=dev, dtype=torch.bfloat16)
    torch_ref.fused_adamw_step(p_ref, g, m_ref, v_ref, 7, 6e-4, 0.9, 0.95,
                               1e-8, 0.1, grad_scale=scale_t,
                               out_bf16=out_ref, commit=True)

    out = torch.empty(n, device=dev, dtype=torch.bfloat16)
    ops.fused_adamw_step(p, g, m, v, 7, 6e-4, 0.9, 0.95, 1e-8, 0.1,
                         grad_scale=scale_t, out_bf16=out, commit=True)
    torch.cuda.synchronize()
    assert torch.allclose(p, p_ref, atol=2e-6, rtol=1e-5)
    assert torch.allclose(m, m_ref, atol=2e-6, rtol=1e-5)
    assert torch.allclose(v, v_ref, atol=2e-6, rtol=1e-5)
    assert (out.float() - out_ref.float()).abs().max() <= 2e-2

    # tentative mode leaves state untouched on GPU too
    p2, m2, v2 = p.clone(), m.clone(), v.clone()
    out_t = torch.empty(n, device=dev, dtype=torch.bfloat16)
    ops.fused_adamw_step(p, g, m, v, 8, 6e-4, 0.9, 0.95, 1e-8, 0.1,
                         grad_scale=0.5, out_bf16=out_t, commit=False)
    torch.cuda.synchronize()
    assert torch.equal(p, p2) and torch.equal(m, m2) and torch.equal(v, v2)
