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
It subsumes the reference's per-projection GEMMs + AccumulateGrad adds on
the flat grad vector (reference trainer_base.py:284-320 re-points
param.grad into the flat tensor; autograd then adds each wgrad into it —
here the wgrad lands in the arena in one step).
"""

from __future__ import annotations

import os
from typing import List, Optional

import torch
import torch.nn as nn

# wgrad accumulation flavor: "1" fuses the += into the hipBLASLt epilogue
# (beta=1); default "0" keeps GEMM-to-temp + vectorized add, measured faster
# on MI355X (hipBLASLt's beta=1 wgrad solutions run ~5-8% slower than
# beta=0, outweighing the saved add kernel — profiles/ r01 v6 vs v7).
_EPILOGUE = os.environ.get("ACCO_WGRAD_EPILOGUE", "0") == "1"

# Grad-producer notifier: when armed (NativeZeroDDP arms it for the last
# micro-batch of an accumulation window), the arena autograd Functions call
# it with each grad-arena view right after accumulating into it, so the DDP
# readiness tracker can launch a bucket's reduce-scatter as soon as every
# producer intersecting the bucket has landed — the in-place-dW equivalent
# of a post-accumulate-grad hook (those never fire for arena views, since
# the weight *view* passed to the Function is not the leaf Parameter).
_grad_notifier = None


def set_grad_notifier(fn) -> None:
    global _grad_notifier
    _grad_notifier = fn


def _notify(g_view) -> None:
    if _grad_notifier is not None:
        _grad_notifier(g_view.storage_offset(), g_view.numel())


def _acc_wgrad(g_view, d2t, x2):
    if _EPILOGUE:
        g_view.addmm_(d2t, x2)
    else:
        g_view.add_(torch.matmul(d2t, x2))
    _notify(g_view)


def _acc_bias_grad(g_b, d2):
    """db += column-sum of d2. Measured same-box: ATen's sum(0)+add wins
    here (GPT-Neo 409.7k vs 361.8k tokens/s with the colsum+cast+add
    chain — the bias case is many small launches, not bandwidth), while
    colsum stays the winner for the norm bwd dW/dB partial reduces inside
    the extension (fewer, larger, fp32). ACCO_COLSUM_BIAS=1 re-enables the
    kernel path for re-measurement."""
    from acco_amd import ops as _ops
    if (d2.is_cuda and _ops.have_kernel("colsum")
            and os.environ.get("ACCO_COLSUM_BIAS") == "1"):
        g_b.add_(_ops.hip_ext().colsum(d2).to(g_b.dtype))
    else:
        g_b.add_(d2.sum(0))


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
        _acc_wgrad(ctx.g_view, d2.t(), x2)
        dx = torch.matmul(dout, w)
        return dx, None, None


class ArenaLinearFn(torch.autograd.Function):
    """Single linear over an arena weight view (optional bias): dW (and db)
    accumulate in-place into the aliased grad-arena views, bypassing
    autograd's temp-dW + AccumulateGrad add."""

    @staticmethod
    def forward(ctx, x, w_view, bias, g_w, g_b):
        ctx.save_for_backward(x, w_view)
        ctx.g_w, ctx.g_b = g_w, g_b
        return torch.nn.functional.linear(x, w_view, bias)

    @staticmethod
    def backward(ctx, dout):
        x, w = ctx.saved_tensors
        d2 = dout.reshape(-1, dout.shape[-1])
        x2 = x.reshape(-1, x.shape[-1])
        _acc_wgrad(ctx.g_w, d2.t(), x2)
        if ctx.g_b is not None:
            _acc_bias_grad(ctx.g_b, d2)
            _notify(ctx.g_b)
        return torch.matmul(dout, w), None, None, None, None


def arena_linear(lin: nn.Linear, x: torch.Tensor) -> torch.Tensor:
    """Run `lin` with in-place grad-arena accumulation if
    `install_fused_projections` wrapped it; plain nn.Linear otherwise."""
    fuse = getattr(lin, "_arena_fuse", None)
    if fuse is None:
        return lin(x)
    w_view, g_w, b, g_b = fuse
    return ArenaLinearFn.apply(x, w_view, b, g_w, g_b)


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


def _param_views(p: nn.Parameter, arena: torch.Tensor,
                 grads: torch.Tensor) -> Optional[tuple]:
    """(weight view, grad view) of `arena`/`grads` aliasing param `p`."""
    if p is None or p.data.untyped_storage().data_ptr() != arena.untyped_storage().data_ptr():
        return None
    off, n = p.data.storage_offset(), p.numel()
    return arena[off:off + n].view(p.shape), grads[off:off + n].view(p.shape)


def _wrap_single(lin: nn.Linear, arena: torch.Tensor,
                 grads: torch.Tensor) -> bool:
    """Give a plain nn.Linear the in-place-dW arena path (`arena_linear`)."""
    got = _param_views(lin.weight, arena, grads)
    if got is None:
        return False
    w_view, g_w = got
    b = g_b = None
    if lin.bias is not None:
        gotb = _param_views(lin.bias, arena, grads)
        if gotb is None:
            return False
        b, g_b = gotb
    lin._arena_fuse = (w_view, g_w, b, g_b)
    return True


def install_fused_projections(model: nn.Module, params_arena: torch.Tensor,
                              grads_arena: torch.Tensor) -> int:
    """Attach fused views to every attention / MLP block whose projections
    are arena-adjacent, and the in-place-dW path to every other arena-resident
    projection (o/down/out/c_fc/c_proj/lm_head). Returns the number of fused
    groups installed."""
    from acco_amd.models.gptneo import GPTNeoSelfAttention
    from acco_amd.models.llama import LlamaAttention, LlamaMLP

    count = 0
    grouped = set()
    for mod in model.modules():
        if isinstance(mod, LlamaAttention):
            got = _arena_views([mod.q_proj.weight, mod.k_proj.weight,
                                mod.v_proj.weight], params_arena, grads_arena)
            if got:
                mod._fused_qkv = got
                grouped.update(id(p) for p in (mod.q_proj.weight,
                                               mod.k_proj.weight,
                                               mod.v_proj.weight))
                count += 1
        elif isinstance(mod, LlamaMLP):
            got = _arena_views([mod.gate_proj.weight, mod.up_proj.weight],
                               params_arena, grads_arena)
            if got:
                mod._fused_gate_up = got
                grouped.update(id(p) for p in (mod.gate_proj.weight,
                                               mod.up_proj.weight))
                count += 1
        elif isinstance(mod, GPTNeoSelfAttention):
            # GPT-Neo declares k, v, q in that order (HF layout)
            got = _arena_views([mod.k_proj.weight, mod.v_proj.weight,
                                mod.q_proj.weight], params_arena, grads_arena)
            if got:
                mod._fused_kvq = got
                grouped.update(id(p) for p in (mod.k_proj.weight,
                                               mod.v_proj.weight,
                                               mod.q_proj.weight))
                count += 1
    for mod in model.modules():
        if (isinstance(mod, nn.Linear) and id(mod.weight) not in grouped
                and mod.weight.dim() == 2):
            _wrap_single(mod, params_arena, grads_arena)
    return count


_FUSE_ATTRS = ("_fused_qkv", "_fused_gate_up", "_fused_kvq", "_arena_fuse")


def snapshot_fused_views(model: nn.Module):
    """Capture the installed arena views (one set per arena for the
    arena-swap optimization: install against each arena, snapshot, then
    flip between snapshots with apply_fused_views — no per-round
    re-derivation)."""
    snap = []
    for mod in model.modules():
        for attr in _FUSE_ATTRS:
            v = getattr(mod, attr, None)
            if v is not None:
                snap.append((mod, attr, v))
    return snap


def apply_fused_views(snapshot) -> None:
    for mod, attr, v in snapshot:
        setattr(mod, attr, v)


def notify_producers(model: nn.Module):
    """Enumerate the grad-arena ranges filled by the in-place-dW notify path
    (fused groups + wrapped singles) as ``[(offset, numel)]``, plus the set
    of Parameter ids those ranges cover. Params NOT covered still produce
    grads through plain autograd (AccumulateGrad); a covered param can
    *additionally* produce through AccumulateGrad when it is also used
    directly (tied lm_head/embedding weight) — callers must count that as a
    separate producer."""
    ranges, covered = [], set()
    for mod in model.modules():
        for attr, names in (("_fused_qkv", ("q_proj", "k_proj", "v_proj")),
                            ("_fused_gate_up", ("gate_proj", "up_proj")),
                            ("_fused_kvq", ("k_proj", "v_proj", "q_proj"))):
            got = getattr(mod, attr, None)
            if got:
                _w, g, _s = got
                ranges.append((g.storage_offset(), g.numel()))
                covered.update(id(getattr(mod, nm).weight) for nm in names)
        af = getattr(mod, "_arena_fuse", None)
        if af is not None:
            _w, g_w, _b, g_b = af
            ranges.append((g_w.storage_offset(), g_w.numel()))
            covered.add(id(mod.weight))
            if g_b is not None:
                ranges.append((g_b.storage_offset(), g_b.numel()))
                covered.add(id(mod.bias))
    return ranges, covered
