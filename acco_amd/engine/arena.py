"""Flat parameter / gradient arenas.

The reference flattens params with parameters_to_vector and then re-points
model tensors at views of the flat vector after the fact
(trainer_base.py:284-332). Here the arenas are first-class: one padded flat
tensor per kind, allocated up front, with every model parameter (and its
.grad) aliasing a contiguous segment. Padding to a communication-friendly
size means collectives (all-gather of params, reduce-scatter of grads) can
target the arena directly with zero staging copies.
"""

from __future__ import annotations

from typing import List

import torch
import torch.nn as nn


def param_order(model: nn.Module) -> List[nn.Parameter]:
    """Deterministic parameter order (module traversal order, deduped —
    tied weights appear once, matching parameters_to_vector semantics)."""
    return list(model.parameters())


def flatten_params(model: nn.Module, dtype: torch.dtype, device,
                   pad_to: int = 1) -> torch.Tensor:
    """Move the model to (device, dtype) and re-home every parameter into a
    single flat arena (padded to a multiple of `pad_to`). Returns the arena;
    arena[:N] is the live parameter vector, arena[N:] is padding."""
    params = param_order(model)
    n = sum(p.numel() for p in params)
    total = ((n + pad_to - 1) // pad_to) * pad_to
    arena = torch.zeros(total, dtype=dtype, device=device)
    off = 0
    with torch.no_grad():
        for p in params:
            num = p.numel()
            seg = arena[off:off + num].view_as(p)
            seg.copy_(p.to(device=device, dtype=dtype))
            p.data = seg
            off += num
    return arena


def attach_grad_arena(model: nn.Module, dtype: torch.dtype, device,
                      pad_to: int = 1) -> torch.Tensor:
    """Allocate a flat gradient arena and point every param.grad at its
    segment; autograd then accumulates in place into the arena
    (the capability of reference trainer_decoupled.py:272-293 prepare_grads,
    without the throwaway forward/backward)."""
    params = param_order(model)
    n = sum(p.numel() for p in params)
    total = ((n + pad_to - 1) // pad_to) * pad_to
    arena = torch.zeros(total, dtype=dtype, device=device)
    off = 0
    for p in params:
        num = p.numel()
        p.grad = arena[off:off + num].view_as(p)
        off += num
    return arena


def repoint_params(model: nn.Module, arena: torch.Tensor) -> None:
    """Re-home every parameter's data view onto `arena` (same flat order as
    flatten_params) WITHOUT copying — the arena-swap optimization: after an
    ACCO com round the com buffer already holds the new parameters, so the
    params-arena and com-buffer swap ROLES instead of moving 2·N bytes."""
    off = 0
    with torch.no_grad():
        for p in param_order(model):
            num = p.numel()
            p.data = arena[off:off + num].view_as(p)
            off += num


def live_numel(model: nn.Module) -> int:
    return sum(p.numel() for p in model.parameters())


def check_aliasing(model: nn.Module, params_arena: torch.Tensor,
                   grads_arena: torch.Tensor | None = None) -> bool:
    """Debug assertion: every param (and grad) storage is the arena's."""
    ps = params_arena.untyped_storage().data_ptr()
    for p in model.parameters():
        if p.data.untyped_storage().data_ptr() != ps:
            return False
        if grads_arena is not None:
            if p.grad is None:
                return False
            if p.grad.untyped_storage().data_ptr() != grads_arena.untyped_storage().data_ptr():
                return False
    return True
