"""On a GPU box the flagship-shape ops MUST dispatch to the HIP autograd
Functions — never silently to the torch reference (whose numerics would
make the parity tests pass vacuously). Asserted via grad_fn identity."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def _fn_name(t: torch.Tensor) -> str:
    return type(t.grad_fn).__name__


@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU")
def test_flagship_ops_dispatch_to_hip():
    from acco_amd import ops
    dev = torch.device("cuda")

    # attention at the llama-1b flagship shape (B,S,H,D layout)
    q = torch.randn(1, 1024, 4, 64, device=dev, dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn_like(q, requires_grad=True)
    v = torch.randn_like(q, requires_grad=True)
    o = ops.causal_attention(q, k, v)
    assert _fn_name(o) == "AttentionFnBackward", _fn_name(o)

    x = torch.randn(64, 2048, device=dev, dtype=torch.bfloat16,
                    requires_grad=True)
    w = torch.ones(2048, device=dev, dtype=torch.bfloat16,
                   requires_grad=True)
    y = ops.rms_norm(x, w, 1e-5)
    assert _fn_name(y) == "RMSNormFnBackward", _fn_name(y)

    res = torch.randn_like(x, requires_grad=True)
    y2, s = ops.add_rms_norm(x, res, w, 1e-5)
    assert _fn_name(y2) == "AddRMSNormFnBackward", _fn_name(y2)

    b = torch.zeros(2048, device=dev, dtype=torch.bfloat16,
                    requires_grad=True)
    y3 = ops.layer_norm(x, w, b, 1e-5)
    assert _fn_name(y3) == "LayerNormFnBackward", _fn_name(y3)

    g = torch.randn(64, 1024, device=dev, dtype=torch.bfloat16,
                    requires_grad=True)
    u = torch.randn_like(g, requires_grad=True)
    y4 = ops.swiglu(g, u)
    assert _fn_name(y4) == "SwiGLUFnBackward", _fn_name(y4)

    y5 = ops.gelu_new(g)
    assert _fn_name(y5) == "GeluNewFnBackward", _fn_name(y5)

    logits = torch.randn(2, 128, 50304, device=dev, dtype=torch.bfloat16,
                         requires_grad=True)
    labels = torch.randint(0, 50304, (2, 128), device=dev)
    loss = ops.causal_lm_loss(logits, labels)
    assert _fn_name(loss) == "CausalLMLossFnBackward", _fn_name(loss)
    loss_ls = ops.label_smoothed_causal_lm_loss(logits, labels, 0.1)
    assert _fn_name(loss_ls) == "CausalLMLossFnBackward", _fn_name(loss_ls)
