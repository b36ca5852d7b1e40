"""Fused projection GEMMs over flat-arena weight views.

The flat parameter arena stores weights in module declaration order, so the
q/k/v projection matrices of an attention block (and gate/up of a SwiGLU
MLP) occupy *adjacent* arena segments. Their concatenation along the output
dimension is therefore already materialized: a single [out_total, in] view
of the arena. `install_fused_projections` verifies adjacency and hands each
block a fused weight view + the matching fused grad-arena view; the block's
forward then issues ONE GEMM instead of 2-3, and the custom autograd
accumulates dW straight into the grad arena (exactly where AccumulateGrad
would have put the per-projection grads).

This is an MI355X-first arena dividend: bigger GEMMs fill the 256-CU chip
(guide: a launch needs >>256 workgroups) and halve kernel-launch count.
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.nn as nn


class FusedArenaLinearFn(torch.autograd.Function):
    """y = x @ W^T with W a flat-arena view; dW accumulates in-place into
    the aliased grad-arena view (bias-free projections only)."""

    @staticmethod
    def forward(ctx, x, w_view, g_view):
        ctx.save_for_backward(x, w_view)
        ctx.g_view = g_view
        return torch.matmul(x, w_view.t())

    @staticmethod
    def backward(ctx, dout):
        x, w = ctx.saved_tensors
        d2 = dout.reshape(-1, dout.shape[-1])
        x2 = x.reshape(-1, x.shape[-1])
        ctx.g_view.add_(torch.matmul(d2.t(), x2))
        dx = torch.matmul(dout, w)
        return dx, None, None


def _arena_views(params: List[nn.Parameter], arena: torch.Tensor,
                 grads: torch.Tensor) -> Optional[tuple]:
    """If the params' storages are consecutive segments of `arena`, return
    (fused weight view, fused grad view, split sizes)."""
    base = arena.untyped_storage().data_ptr()
    esz = arena.element_size()
    offs = []
    for p in params:
        if p.data.untyped_storage().data_ptr() != base:
            return None
        offs.append((p.data.storage_offset(), p.numel(), p.shape))
    offs_sorted = sorted(offs)
    if offs_sorted != offs:
        return None
    start = offs[0][0]
    cur = start
    in_dim = offs[0][2][1]
    out_total = 0
    splits = []
    for off, n, shape in offs:
        if off != cur or len(shape) != 2 or shape[1] != in_dim:
            return None
        cur += n
        out_total += shape[0]
        splits.append(shape[0])
    w_view = arena[start:cur].view(out_total, in_dim)
    g_view = grads[start:cur].view(out_total, in_dim)
    del esz
    return w_view, g_view, splits


def install_fused_projections(model: nn.Module, params_arena: torch.Tensor,
                              grads_arena: torch.Tensor) -> int:
    """Attach fused views to every attention / MLP block whose projections
    are arena-adjacent. Returns the number of fused groups installed."""
    from acco_amd.models.gptneo import GPTNeoSelfAttention
    from acco_amd.models.llama import LlamaAttention, LlamaMLP

    count = 0
    for mod in model.modules():
        if isinstance(mod, LlamaAttention):
            got = _arena_views([mod.q_proj.weight, mod.k_proj.weight,
                                mod.v_proj.weight], params_arena, grads_arena)
            if got:
                mod._fused_qkv = got
                count += 1
        elif isinstance(mod, LlamaMLP):
            got = _arena_views([mod.gate_proj.weight, mod.up_proj.weight],
                               params_arena, grads_arena)
            if got:
                mod._fused_gate_up = got
                count += 1
        elif isinstance(mod, GPTNeoSelfAttention):
            # GPT-Neo declares k, v, q in that order (HF layout)
            got = _arena_views([mod.k_proj.weight, mod.v_proj.weight,
                                mod.q_proj.weight], params_arena, grads_arena)
            if got:
                mod._fused_kvq = got
                count += 1
    return count
