"""Llama-family causal LM, MI355X-native.

A from-scratch module whose ``state_dict`` keys match HF
``LlamaForCausalLM`` (model.embed_tokens / model.layers.N.self_attn.q_proj /
mlp.gate_proj / input_layernorm / model.norm / lm_head), so HF checkpoints
load directly (checkpoint-layout parity with the reference, which saves HF
``state_dict`` — reference trainer_decoupled.py:559-574).

All hot ops route through acco_amd.ops: RMSNorm, RoPE, SwiGLU, causal flash
attention and the CE loss are hand-written gfx950 HIP kernels on GPU
(SURVEY.md §2.5 K1/K2 rebuild); GEMMs go to hipBLASLt via torch.nn.Linear.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from acco_amd import ops
from acco_amd.models.config import LlamaConfig
from acco_amd.models.fuse import arena_linear


class LlamaRMSNorm(nn.Module):
    def __init__(self, hidden_size: int, eps: float):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden_size))
        self.variance_epsilon = eps

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return ops.rms_norm(x, self.weight, self.variance_epsilon)


class LlamaAttention(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        d, hd = cfg.hidden_size, cfg.head_dim
        self.q_proj = nn.Linear(d, cfg.num_heads * hd, bias=False)
        self.k_proj = nn.Linear(d, cfg.num_kv_heads * hd, bias=False)
        self.v_proj = nn.Linear(d, cfg.num_kv_heads * hd, bias=False)
        self.o_proj = nn.Linear(cfg.num_heads * hd, d, bias=False)
        self._fused_qkv = None    # (w_view, g_view, splits) via models.fuse

    def forward(self, x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
        B, S, _ = x.shape
        cfg = self.cfg
        # keep everything in the projections' natural [B, S, H, D] layout:
        # RoPE and attention kernels walk D coalesced, no transpose copies
        if self._fused_qkv is not None:
            from acco_amd.models.fuse import FusedArenaLinearFn
            w, g, splits = self._fused_qkv
            qkv = FusedArenaLinearFn.apply(x, w, g)
            if (qkv.is_cuda and cfg.head_dim in (64, 128) and S % 256 == 0
                    and ops.have_kernel("attn_fwd_packed")):
                # packed core: RoPE + flash attention directly on the fused
                # projection output, packed grad back — zero split/cat
                from acco_amd.ops.autograd import AttnQKVPackedFn
                hd = cfg.head_dim
                offs = (0, cfg.num_heads * hd,
                        (cfg.num_heads + cfg.num_kv_heads) * hd)
                o = AttnQKVPackedFn.apply(qkv, cos, sin, cfg.num_heads,
                                          cfg.num_kv_heads, hd,
                                          hd ** -0.5, 0, offs)
                return arena_linear(self.o_proj, o)
            q, k, v = torch.split(qkv, splits, dim=-1)
            q = q.contiguous().view(B, S, cfg.num_heads, cfg.head_dim)
            k = k.contiguous().view(B, S, cfg.num_kv_heads, cfg.head_dim)
            v = v.contiguous().view(B, S, cfg.num_kv_heads, cfg.head_dim)
        else:
            q = self.q_proj(x).view(B, S, cfg.num_heads, cfg.head_dim)
            k = self.k_proj(x).view(B, S, cfg.num_kv_heads, cfg.head_dim)
            v = self.v_proj(x).view(B, S, cfg.num_kv_heads, cfg.head_dim)
        q, k = ops.rope_apply(q, k, cos, sin)
        o = ops.causal_attention(q, k, v)          # [B, S, H, D]
        return arena_linear(self.o_proj, o.reshape(B, S, -1))


class LlamaMLP(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.gate_proj = nn.Linear(cfg.hidden_size, cfg.intermediate_size, bias=False)
        self.up_proj = nn.Linear(cfg.hidden_size, cfg.intermediate_size, bias=False)
        self.down_proj = nn.Linear(cfg.intermediate_size, cfg.hidden_size, bias=False)
        self._fused_gate_up = None

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self._fused_gate_up is not None:
            from acco_amd.models.fuse import FusedArenaLinearFn
            w, g, splits = self._fused_gate_up
            gu = FusedArenaLinearFn.apply(x, w, g)
            if gu.is_cuda and ops.have_kernel("swiglu_packed_fwd"):
                from acco_amd.ops.autograd import SwiGLUPackedFn
                return arena_linear(self.down_proj, SwiGLUPackedFn.apply(gu))
            gate, up = torch.split(gu, splits, dim=-1)
            return arena_linear(self.down_proj, ops.swiglu(gate.contiguous(),
                                             up.contiguous()))
        return arena_linear(self.down_proj, ops.swiglu(self.gate_proj(x), self.up_proj(x)))


class LlamaDecoderLayer(nn.Module):
    """Pre-norm decoder layer in (pending, residual) form: every residual
    add is FUSED into the following norm kernel (ops.add_rms_norm returns
    both the normed value and the sum). The layer receives the previous
    layer's un-added MLP output (`pending`) plus the running residual, and
    returns its own un-added MLP output — the model's final norm completes
    the last add. Equivalent math to the classic
    x = x + attn(norm1(x)); x = x + mlp(norm2(x)) chain."""

    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.self_attn = LlamaAttention(cfg)
        self.mlp = LlamaMLP(cfg)
        self.input_layernorm = LlamaRMSNorm(cfg.hidden_size, cfg.rms_norm_eps)
        self.post_attention_layernorm = LlamaRMSNorm(cfg.hidden_size, cfg.rms_norm_eps)

    def forward(self, pending, residual, cos, sin):
        n1 = self.input_layernorm
        h1, residual = ops.add_rms_norm(pending, residual, n1.weight,
                                        n1.variance_epsilon)
        a = self.self_attn(h1, cos, sin)
        n2 = self.post_attention_layernorm
        h2, residual = ops.add_rms_norm(a, residual, n2.weight,
                                        n2.variance_epsilon)
        return self.mlp(h2), residual


class LlamaModel(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(LlamaDecoderLayer(cfg) for _ in range(cfg.num_layers))
        self.norm = LlamaRMSNorm(cfg.hidden_size, cfg.rms_norm_eps)
        # RoPE tables are buffers recomputed lazily per device/seq-len
        # (host-precomputed trig per guide Appendix B: no on-device sinf/cosf).
        self._rope_cache: dict = {}
        # per-layer activation recompute (train.activation_checkpointing —
        # 8B-on-288GB batch headroom); deterministic
        # forward (no dropout), so plain non-reentrant checkpointing is exact
        self.gradient_checkpointing = False

    def _cos_sin(self, S: int, device):
        key = (S, device)
        hit = self._rope_cache.get(key)
        if hit is None:
            # fp32 tables: the RoPE HIP kernel multiplies in fp32
            hit = ops.torch_ref.rope_cos_sin(S, self.cfg.head_dim,
                                             self.cfg.rope_theta, device)
            self._rope_cache = {key: hit}   # keep one entry (static shapes)
        return hit

    def forward(self, input_ids: torch.Tensor) -> torch.Tensor:
        pending = self.embed_tokens(input_ids)
        residual = None
        cos, sin = self._cos_sin(input_ids.shape[1], pending.device)
        if (self.gradient_checkpointing and self.training
                and torch.is_grad_enabled()):
            from torch.utils.checkpoint import checkpoint
            for layer in self.layers:
                pending, residual = checkpoint(layer, pending, residual,
                                               cos, sin, use_reentrant=False)
        else:
            for layer in self.layers:
                pending, residual = layer(pending, residual, cos, sin)
        # final norm completes the last residual add in the same kernel
        y, _ = ops.add_rms_norm(pending, residual, self.norm.weight,
                                self.norm.variance_epsilon)
        return y


class LlamaForCausalLM(nn.Module):
    config_class = LlamaConfig

    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        self.model = LlamaModel(cfg)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight
        self.apply(self._init_weights)

    def _init_weights(self, module):
        std = self.cfg.initializer_range
        if isinstance(module, nn.Linear):
            module.weight.data.normal_(0.0, std)
            if module.bias is not None:
                module.bias.data.zero_()
        elif isinstance(module, nn.Embedding):
            module.weight.data.normal_(0.0, std)

    def forward(self, input_ids: torch.Tensor,
                labels: Optional[torch.Tensor] = None,
                attention_mask: Optional[torch.Tensor] = None):
        """HF-compatible call: returns (loss, logits) when labels given,
        (logits,) otherwise — the trainer indexes outputs[0]
        (reference trainer_decoupled.py:29-34)."""
        hidden = self.model(input_ids)
        logits = arena_linear(self.lm_head, hidden)
        if labels is None:
            return (logits,)
        loss = ops.causal_lm_loss(logits, labels)
        return (loss, logits)
