"""Pure-PyTorch reference implementations of every hot op.

These are (a) the CPU execution path for tests (this container has no GPU),
and (b) the numerics oracle the gfx950 HIP kernels are validated against
(tests compare HIP output to these run in fp32).

Each docstring cites the reference call-site whose computation the op
replaces (SURVEY.md §2.5 kernel inventory).
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.nn.functional as F


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    """RMSNorm (Llama-family; replaces HF modeling_llama RMSNorm inside K1,
    reference trainer_decoupled.py:26-34 forward)."""
    dt = x.dtype
    xf = x.float()
    var = xf.pow(2).mean(-1, keepdim=True)
    return (xf * torch.rsqrt(var + eps)).to(dt) * weight


def layer_norm(x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor,
               eps: float) -> torch.Tensor:
    """LayerNorm (GPT-Neo family, K1)."""
    return F.layer_norm(x, (x.shape[-1],), weight, bias, eps)


def gelu_new(x: torch.Tensor) -> torch.Tensor:
    """gelu_new / tanh-approximated GELU (GPT-Neo MLP activation,
    reference config/model/gpt-neo-125M.json:2)."""
    return 0.5 * x * (1.0 + torch.tanh(
        math.sqrt(2.0 / math.pi) * (x + 0.044715 * torch.pow(x, 3.0))))


def swiglu(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    """SwiGLU: silu(gate) * up (Llama MLP, K1)."""
    return F.silu(gate) * up


def rope_cos_sin(seq_len: int, head_dim: int, theta: float,
                 device, dtype=torch.float32) -> Tuple[torch.Tensor, torch.Tensor]:
    """Precomputed RoPE tables, HF-Llama layout: [S, D] with the two halves
    duplicated (cos = cat(c, c), sin = cat(s, s))."""
    inv_freq = 1.0 / (theta ** (torch.arange(0, head_dim, 2, device=device,
                                             dtype=torch.float32) / head_dim))
    t = torch.arange(seq_len, device=device, dtype=torch.float32)
    freqs = torch.outer(t, inv_freq)          # [S, D/2]
    emb = torch.cat((freqs, freqs), dim=-1)   # [S, D]
    return emb.cos().to(dtype), emb.sin().to(dtype)


def _rotate_half(x: torch.Tensor) -> torch.Tensor:
    half = x.shape[-1] // 2
    return torch.cat((-x[..., half:], x[..., :half]), dim=-1)


def rope_apply(q: torch.Tensor, k: torch.Tensor, cos: torch.Tensor,
               sin: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """Apply rotary embedding, HF-Llama rotate_half convention.

    q: [B, S, H, D]; k: [B, S, Hkv, D] (the projection's natural contiguous
    layout — the MI355X kernel walks D coalesced with no transpose copies);
    cos/sin: [S, D]."""
    cos = cos.to(q.dtype)[None, :, None, :]
    sin = sin.to(q.dtype)[None, :, None, :]
    q2 = q * cos + _rotate_half(q) * sin
    k2 = k * cos + _rotate_half(k) * sin
    return q2, k2


def causal_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                     scale: Optional[float] = None,
                     window: Optional[int] = None) -> torch.Tensor:
    """Causal (optionally banded/local) attention.

    q: [B, S, H, D]; k, v: [B, S, Hkv, D] (GQA: H a multiple of Hkv) —
    the projections' natural contiguous layout. Returns [B, S, H, D].
    `scale=None` → 1/sqrt(D); GPT-Neo passes scale=1.0 (it does not scale).
    `window` (GPT-Neo local layers, window_size=256): query i attends to
    keys j with i-window < j <= i.
    """
    B, S, H, D = q.shape
    Hkv = k.shape[2]
    q = q.transpose(1, 2)
    k = k.transpose(1, 2)
    v = v.transpose(1, 2)
    if Hkv != H:
        rep = H // Hkv
        k = k.repeat_interleave(rep, dim=1)
        v = v.repeat_interleave(rep, dim=1)
    if scale is None:
        scale = 1.0 / math.sqrt(D)
    scores = torch.matmul(q.float(), k.float().transpose(-1, -2)) * scale
    idx = torch.arange(S, device=q.device)
    mask = idx[None, :] > idx[:, None]           # future → masked
    if window is not None:
        mask = mask | (idx[None, :] <= idx[:, None] - window)
    scores = scores.masked_fill(mask, float("-inf"))
    probs = torch.softmax(scores, dim=-1)
    out = torch.matmul(probs, v.float()).to(q.dtype)
    return out.transpose(1, 2)


def causal_lm_loss(logits: torch.Tensor, labels: torch.Tensor,
                   ignore_index: int = -100) -> torch.Tensor:
    """Shifted causal-LM cross entropy (HF CausalLM loss inside K1/K9)."""
    shift_logits = logits[..., :-1, :].contiguous().float()
    shift_labels = labels[..., 1:].contiguous()
    return F.cross_entropy(shift_logits.view(-1, shift_logits.size(-1)),
                           shift_labels.view(-1), ignore_index=ignore_index)


def label_smoothed_causal_lm_loss(logits: torch.Tensor, labels: torch.Tensor,
                                  epsilon: float,
                                  ignore_index: int = -100) -> torch.Tensor:
    """HF LabelSmoother-equivalent shifted loss
    (reference utils/trainer_utils.py:862-902, used when
    label_smoothing_factor != 0 — trainer_base.py:63-68):
    (1-eps)·NLL + eps·(mean over vocab of -log p), masked mean."""
    shift_logits = logits[..., :-1, :].contiguous().float()
    shift_labels = labels[..., 1:].contiguous()
    logp = F.log_softmax(shift_logits, dim=-1)
    mask = shift_labels.eq(ignore_index)
    safe = shift_labels.clamp(min=0).unsqueeze(-1)
    nll = -logp.gather(-1, safe).squeeze(-1).masked_fill(mask, 0.0)
    smooth = -logp.sum(-1).masked_fill(mask, 0.0)
    n = (~mask).sum().clamp(min=1)
    nll = nll.sum() / n
    smooth = smooth.sum() / (n * shift_logits.size(-1))
    return (1.0 - epsilon) * nll + epsilon * smooth


@torch.no_grad()
def fused_adamw_step(p: torch.Tensor, g: torch.Tensor, m: torch.Tensor,
                     v: torch.Tensor, step: int, lr: float, beta1: float,
                     beta2: float, eps: float, weight_decay: float,
                     grad_scale: float = 1.0,
                     out_bf16: Optional[torch.Tensor] = None,
                     commit: bool = True) -> None:
    """Sharded AdamW step, torch.optim.AdamW-equivalent math (K3+K5 fused,
    reference trainer_decoupled.py:95-100).

    p, m, v fp32; g any float dtype (cast + scaled by `grad_scale` = 1/count,
    reference :98). If `commit` is False this is the ACCO *tentative* step
    (even com rounds, reference :79-84,113-125): the updated parameters are
    written to `out_bf16` but p/m/v/step are left untouched — equivalent to
    the reference's snapshot→step→rollback without the three state clones.
    """
    gf = g.float() * grad_scale
    t = step + 1
    bc1 = 1.0 - beta1 ** t
    bc2 = 1.0 - beta2 ** t
    if commit:
        p.mul_(1.0 - lr * weight_decay)
        m.mul_(beta1).add_(gf, alpha=1.0 - beta1)
        v.mul_(beta2).addcmul_(gf, gf, value=1.0 - beta2)
        denom = (v.sqrt() / math.sqrt(bc2)).add_(eps)
        p.addcdiv_(m, denom, value=-lr / bc1)
        if out_bf16 is not None:
            out_bf16.copy_(p)
    else:
        p_t = p * (1.0 - lr * weight_decay)
        m_t = m * beta1 + gf * (1.0 - beta1)
        v_t = v * beta2 + gf * gf * (1.0 - beta2)
        denom = (v_t.sqrt() / math.sqrt(bc2)).add_(eps)
        p_t.addcdiv_(m_t, denom, value=-lr / bc1)
        assert out_bf16 is not None, "tentative step must write out_bf16"
        out_bf16.copy_(p_t)
