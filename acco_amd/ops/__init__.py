"""Op dispatch: hand-written gfx950 HIP kernels on GPU, torch reference on CPU.

Policy (deliberate, per the MI355X-native mandate): when a tensor is on a
CUDA/HIP device, the op MUST run through the in-tree HIP extension
``acco_amd._hip_ops``; if the extension is missing we raise instead of
silently falling back to ATen. ``ACCO_FORCE_REF=1`` overrides for A/B
numerics debugging only.
"""

from __future__ import annotations

import os
from typing import Optional, Tuple

import torch

from acco_amd.ops import torch_ref

_EXT = None
_EXT_ERR: Optional[str] = None


def _check_fresh(mod) -> None:
    """Warn if the in-tree .so predates any kernel source (a stale binary
    silently satisfies the import and defeats the fail-loud policy —
    round-1 advisor finding)."""
    import glob
    import warnings
    so = getattr(mod, "__file__", None)
    if not so or not os.path.exists(so):
        return
    so_mtime = os.path.getmtime(so)
    csrc = os.path.join(os.path.dirname(so), "ops", "csrc")
    stale = [os.path.basename(f)
             for f in glob.glob(os.path.join(csrc, "*.hip"))
             + glob.glob(os.path.join(csrc, "*.cpp"))
             if not f.endswith("_hip.hip") and os.path.getmtime(f) > so_mtime]
    if stale:
        warnings.warn(
            f"acco_amd._hip_ops is OLDER than kernel sources {stale}; "
            "rebuild with `python setup.py build_ext --inplace`",
            RuntimeWarning, stacklevel=3)


def _load_ext():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        from acco_amd import _hip_ops  # built by setup.py build_ext --inplace
        _check_fresh(_hip_ops)
        _EXT = _hip_ops
    except ImportError as e:  # remember why, for the loud failure below
        _EXT_ERR = str(e)
    return _EXT


def hip_ext():
    """The HIP extension module, or a loud failure on a GPU box."""
    ext = _load_ext()
    if ext is None:
        raise RuntimeError(
            "acco_amd._hip_ops is not built but a CUDA tensor reached an op. "
            "Build it in-tree with `python setup.py build_ext --inplace` "
            f"(import error: {_EXT_ERR})"
        )
    return ext


def ext_available() -> bool:
    return _load_ext() is not None


def have_kernel(name: str) -> bool:
    ext = _load_ext()
    return ext is not None and hasattr(ext, name)


def _use_ref(t: torch.Tensor, kernel: str = "") -> bool:
    """torch-reference path when: CPU tensor, forced via env, or the HIP
    kernel for this op has not been built yet. On a GPU box with the
    extension entirely missing, ops fail loudly instead (hip_ext raises) —
    the kernel registry only tolerates per-op gaps during bring-up."""
    if not t.is_cuda:
        return True
    if os.environ.get("ACCO_FORCE_REF") == "1":
        return True
    if kernel:
        if not ext_available():
            hip_ext()  # raises with the build instructions
        return not have_kernel(kernel)
    return False


# ---------------------------------------------------------------- model ops

def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    if _use_ref(x, "rmsnorm_fwd"):
        return torch_ref.rms_norm(x, weight, eps)
    from acco_amd.ops.autograd import RMSNormFn
    return RMSNormFn.apply(x, weight, eps)


def layer_norm(x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor,
               eps: float) -> torch.Tensor:
    if _use_ref(x, "layernorm_fwd"):
        return torch_ref.layer_norm(x, weight, bias, eps)
    from acco_amd.ops.autograd import LayerNormFn
    return LayerNormFn.apply(x, weight, bias, eps)


def add_rms_norm(x: torch.Tensor, res: Optional[torch.Tensor],
                 weight: torch.Tensor, eps: float):
    """(rmsnorm(x+res)·w, x+res) in ONE kernel pass (residual-add fusion);
    res=None degenerates to a plain norm with s = x.
    ACCO_NO_ADDNORM_FUSE=1 keeps the eager add + plain norm (same-box A/B)."""
    if res is None:
        return rms_norm(x, weight, eps), x
    if os.environ.get("ACCO_NO_ADDNORM_FUSE") == "1":
        s = x + res
        return rms_norm(s, weight, eps), s
    if _use_ref(x, "add_rmsnorm_fwd"):
        s = x + res
        return torch_ref.rms_norm(s, weight, eps), s
    from acco_amd.ops.autograd import AddRMSNormFn
    return AddRMSNormFn.apply(x, res, weight, eps)


def add_layer_norm(x: torch.Tensor, res: Optional[torch.Tensor],
                   weight: torch.Tensor, bias: torch.Tensor, eps: float):
    """(layernorm(x+res)·w+b, x+res) in ONE kernel pass; res=None → plain."""
    if res is None:
        return layer_norm(x, weight, bias, eps), x
    if os.environ.get("ACCO_NO_ADDNORM_FUSE") == "1":
        s = x + res
        return layer_norm(s, weight, bias, eps), s
    if _use_ref(x, "add_layernorm_fwd"):
        s = x + res
        return torch_ref.layer_norm(s, weight, bias, eps), s
    from acco_amd.ops.autograd import AddLayerNormFn
    return AddLayerNormFn.apply(x, res, weight, bias, eps)


def gelu_new(x: torch.Tensor) -> torch.Tensor:
    if _use_ref(x, "gelu_fwd"):
        return torch_ref.gelu_new(x)
    from acco_amd.ops.autograd import GeluNewFn
    return GeluNewFn.apply(x)


def swiglu(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    if _use_ref(gate, "swiglu_fwd"):
        return torch_ref.swiglu(gate, up)
    from acco_amd.ops.autograd import SwiGLUFn
    return SwiGLUFn.apply(gate, up)


def rope_apply(q: torch.Tensor, k: torch.Tensor, cos: torch.Tensor,
               sin: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    if _use_ref(q, "rope_fwd"):
        return torch_ref.rope_apply(q, k, cos, sin)
    from acco_amd.ops.autograd import RoPEFn
    return RoPEFn.apply(q, k, cos, sin)


def causal_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                     scale: Optional[float] = None,
                     window: Optional[int] = None) -> torch.Tensor:
    if _use_ref(q, "attn_fwd"):
        return torch_ref.causal_attention(q, k, v, scale=scale, window=window)
    S, D = q.shape[1], q.shape[3]
    if S % 64 != 0 or D not in (64, 128) or q.dtype != torch.bfloat16:
        # shapes outside the flash kernel's contract (rare: tests, odd
        # eval batches) run the reference math
        return torch_ref.causal_attention(q, k, v, scale=scale, window=window)
    from acco_amd.ops.autograd import AttentionFn
    return AttentionFn.apply(q, k, v, scale, window)


def causal_lm_loss(logits: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
    if _use_ref(logits, "ce_fwd"):
        return torch_ref.causal_lm_loss(logits, labels)
    from acco_amd.ops.autograd import CausalLMLossFn
    return CausalLMLossFn.apply(logits, labels)


def label_smoothed_causal_lm_loss(logits: torch.Tensor, labels: torch.Tensor,
                                  epsilon: float) -> torch.Tensor:
    """HF-LabelSmoother-equivalent shifted loss fused into the CE kernel
    (reference utils/trainer_utils.py:862-902)."""
    if epsilon == 0.0:
        return causal_lm_loss(logits, labels)
    if _use_ref(logits, "ce_fwd"):
        return torch_ref.label_smoothed_causal_lm_loss(logits, labels, epsilon)
    from acco_amd.ops.autograd import CausalLMLossFn
    return CausalLMLossFn.apply(logits, labels, epsilon)


# ------------------------------------------------------------- trainer ops

def fused_adamw_step(p: torch.Tensor, g: torch.Tensor, m: torch.Tensor,
                     v: torch.Tensor, step: int, lr: float, beta1: float,
                     beta2: float, eps: float, weight_decay: float,
                     grad_scale: float = 1.0,
                     out_bf16: Optional[torch.Tensor] = None,
                     commit: bool = True) -> None:
    """Sharded AdamW on the local fp32 shard, with bf16-grad cast, 1/count
    scale, optional bf16 write-out, and the ACCO tentative (no-commit) mode.
    One HIP kernel on GPU (K3+K4+K5+K7 of SURVEY.md §2.5 fused)."""
    if _use_ref(p):
        torch_ref.fused_adamw_step(p, g, m, v, step, lr, beta1, beta2, eps,
                                   weight_decay, grad_scale, out_bf16, commit)
        return
    if isinstance(grad_scale, torch.Tensor):
        scale, scale_dev = 1.0, grad_scale.float()
    else:
        scale = float(grad_scale)
        scale_dev = torch.empty(0, device=p.device, dtype=torch.float32)
    if out_bf16 is None:
        out_bf16 = torch.empty(0, device=p.device, dtype=g.dtype)
    hip_ext().fused_adamw(p, g, m, v, int(step), float(lr), float(beta1),
                          float(beta2), float(eps), float(weight_decay),
                          scale, scale_dev, out_bf16, bool(commit))
