"""GPT-Neo-family causal LM, MI355X-native.

From-scratch module with ``state_dict`` keys matching HF
``GPTNeoForCausalLM`` (transformer.wte/wpe, transformer.h.N.ln_1,
attn.attention.{q,k,v,out}_proj, mlp.c_fc/c_proj, transformer.ln_f,
lm_head tied to wte), so HF checkpoints load directly.

GPT-Neo specifics faithfully kept (reference config/model/gpt-neo-125M.json):
- learned absolute position embeddings (wpe);
- alternating global / local (banded, window 256) causal attention;
- NO attention scaling (scale = 1.0 — GPT-Neo quirk);
- gelu_new MLP activation; q/k/v projections have no bias, out_proj does.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from acco_amd import ops
from acco_amd.models.fuse import arena_linear
from acco_amd.models.config import GPTNeoConfig


class GPTNeoSelfAttention(nn.Module):
    def __init__(self, cfg: GPTNeoConfig, attention_type: str):
        super().__init__()
        d = cfg.hidden_size
        self.cfg = cfg
        self.attention_type = attention_type
        self.k_proj = nn.Linear(d, d, bias=False)
        self.v_proj = nn.Linear(d, d, bias=False)
        self.q_proj = nn.Linear(d, d, bias=False)
        self.out_proj = nn.Linear(d, d, bias=True)
        self._fused_kvq = None    # (w_view, g_view, splits) via models.fuse

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, S, d = x.shape
        H, hd = self.cfg.num_heads, self.cfg.head_dim
        if self._fused_kvq is not None:
            from acco_amd import ops as _ops
            from acco_amd.models.fuse import FusedArenaLinearFn
            w, g, splits = self._fused_kvq
            kvq = FusedArenaLinearFn.apply(x, w, g)
            window = (self.cfg.window_size
                      if self.attention_type == "local" else 0)
            if (kvq.is_cuda and hd == 64 and S % 256 == 0
                    and _ops.have_kernel("attn_fwd_packed")):
                from acco_amd.ops.autograd import AttnQKVPackedFn
                # GPT-Neo packing order k|v|q; no RoPE; no 1/sqrt(d) scale
                offs = (2 * d, 0, d)
                o = AttnQKVPackedFn.apply(kvq, None, None, H, H, hd, 1.0,
                                          window, offs)
                return arena_linear(self.out_proj, o)
            k, v, q = torch.split(kvq, splits, dim=-1)
            q = q.contiguous().view(B, S, H, hd)
            k = k.contiguous().view(B, S, H, hd)
            v = v.contiguous().view(B, S, H, hd)
        else:
            q = self.q_proj(x).view(B, S, H, hd)
            k = self.k_proj(x).view(B, S, H, hd)
            v = self.v_proj(x).view(B, S, H, hd)
        window = self.cfg.window_size if self.attention_type == "local" else None
        o = ops.causal_attention(q, k, v, scale=1.0, window=window)
        return arena_linear(self.out_proj, o.reshape(B, S, d))


class GPTNeoAttention(nn.Module):
    """Wrapper level kept so state_dict keys read attn.attention.* like HF."""

    def __init__(self, cfg: GPTNeoConfig, layer_id: int):
        super().__init__()
        self.attention = GPTNeoSelfAttention(cfg, cfg.layer_attention_type(layer_id))

    def forward(self, x):
        return self.attention(x)


class GPTNeoMLP(nn.Module):
    def __init__(self, cfg: GPTNeoConfig):
        super().__init__()
        self.c_fc = nn.Linear(cfg.hidden_size, cfg.inner_size, bias=True)
        self.c_proj = nn.Linear(cfg.inner_size, cfg.hidden_size, bias=True)

    def forward(self, x):
        return arena_linear(self.c_proj, ops.gelu_new(arena_linear(self.c_fc, x)))


class GPTNeoBlock(nn.Module):
    def __init__(self, cfg: GPTNeoConfig, layer_id: int):
        super().__init__()
        self.ln_1 = nn.LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_epsilon)
        self.attn = GPTNeoAttention(cfg, layer_id)
        self.ln_2 = nn.LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_epsilon)
        self.mlp = GPTNeoMLP(cfg)

    def forward(self, pending, residual):
        """(pending, residual) form — every residual add fused into the
        following LayerNorm kernel (see LlamaDecoderLayer); returns the
        un-added MLP output + running residual, ln_f completes the last add."""
        h1, residual = ops.add_layer_norm(pending, residual, self.ln_1.weight,
                                          self.ln_1.bias, self.ln_1.eps)
        a = self.attn(h1)
        h2, residual = ops.add_layer_norm(a, residual, self.ln_2.weight,
                                          self.ln_2.bias, self.ln_2.eps)
        return self.mlp(h2), residual


class GPTNeoModel(nn.Module):
    def __init__(self, cfg: GPTNeoConfig):
        super().__init__()
        self.cfg = cfg
        self.wte = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.wpe = nn.Embedding(cfg.max_position_embeddings, cfg.hidden_size)
        self.h = nn.ModuleList(GPTNeoBlock(cfg, i) for i in range(cfg.num_layers))
        self.ln_f = nn.LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_epsilon)
        # per-layer activation recompute (train.activation_checkpointing)
        self.gradient_checkpointing = False

    def forward(self, input_ids: torch.Tensor) -> torch.Tensor:
        S = input_ids.shape[1]
        pos = torch.arange(S, device=input_ids.device)
        pending = self.wte(input_ids) + self.wpe(pos)[None]
        residual = None
        if (self.gradient_checkpointing and self.training
                and torch.is_grad_enabled()):
            from torch.utils.checkpoint import checkpoint
            for block in self.h:
                pending, residual = checkpoint(block, pending, residual,
                                               use_reentrant=False)
        else:
            for block in self.h:
                pending, residual = block(pending, residual)
        y, _ = ops.add_layer_norm(pending, residual, self.ln_f.weight,
                                  self.ln_f.bias, self.ln_f.eps)
        return y


class GPTNeoForCausalLM(nn.Module):
    config_class = GPTNeoConfig

    def __init__(self, cfg: GPTNeoConfig):
        super().__init__()
        self.cfg = cfg
        self.transformer = GPTNeoModel(cfg)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.transformer.wte.weight
        self.apply(self._init_weights)

    def _init_weights(self, module):
        std = self.cfg.initializer_range
        if isinstance(module, nn.Linear):
            module.weight.data.normal_(0.0, std)
            if module.bias is not None:
                module.bias.data.zero_()
        elif isinstance(module, nn.Embedding):
            module.weight.data.normal_(0.0, std)
        elif isinstance(module, nn.LayerNorm):
            module.weight.data.fill_(1.0)
            module.bias.data.zero_()

    def forward(self, input_ids: torch.Tensor,
                labels: Optional[torch.Tensor] = None,
                attention_mask: Optional[torch.Tensor] = None):
        hidden = self.transformer(input_ids)
        logits = arena_linear(self.lm_head, hidden)
        if labels is None:
            return (logits,)
        loss = ops.causal_lm_loss(logits, labels)
        return (loss, logits)
