"""Native synchronous DDP baseline with ZeRO-1 sharded AdamW.

MI355X-native replacement for the reference's
``DistributedDataParallel`` + ``ZeroRedundancyOptimizer`` pair
(trainer_decoupled.py:226-241, 732-763), built on the same flat arenas,
bucket-major ShardSpec and fused AdamW kernel as the ACCO engine:

- gradients accumulate in the flat grad arena; on the *last* micro-batch of
  an accumulation window, per-bucket reduce-scatter launches as soon as the
  bucket's gradients are complete (overlapped with the rest of backward,
  like DDP's bucketed all-reduce);
- each rank then runs the fused AdamW on its own (bucket, rank) segments of
  the grad arena and writes updated bf16 params straight into the *params
  arena* segment, which the per-bucket all-gather completes in place —
  model weights update with zero staging copies.

Bucket readiness is tracked by ELEMENT COVERAGE, not per-param hook counts:
every gradient producer (a plain param's post-accumulate-grad hook, a fused
qkv/gate-up group, or a wrapped single linear whose dW lands in the arena
in place via models.fuse) contributes its arena range; a bucket launches
when the covered element count reaches the bucket's requirement. This is
what lets the DDP baseline run the SAME fused arena-view projections as the
ACCO engine (round-1 gated them out of DDP, handicapping the baseline —
the ACCO-vs-DDP comparison must measure the algorithm, not a kernel gap).
Tied weights (lm_head aliasing embed_tokens) are two producers for the same
range — the in-place lm_head dW and the embedding's AccumulateGrad — and
the coverage count includes both, so the bucket never launches before the
embedding backward lands.

This moves strictly fewer bytes than the reference baseline
(reduce-scatter + all-gather vs all-reduce + shard broadcast) on the
per-link-bound xGMI fabric.

Cross-rank collective ordering (the RCCL/NCCL hazard): collectives on one
communicator must be issued in the same order on every rank or the ranks
can deadlock. Overlapped bucket launches are safe here because autograd
uses ONE worker thread per device and every rank executes the identical
backward graph, so producers finish — and coverage thresholds trip — in
the same deterministic order on all ranks; the lock only guards the
counter arithmetic, not ordering. Buckets never launched in backward
(padding-only or no-grad buckets) are launched by ``finish_step`` in
ascending bucket order, again identical on all ranks.
"""

from __future__ import annotations

import threading
from typing import List, Optional, Union

import torch
import torch.nn as nn

from acco_amd.engine.sharded_adamw import ShardedAdamW
from acco_amd.models import fuse as fuse_mod
from acco_amd.parallel.comm import CommBackend, ShardSpec


class NativeZeroDDP:
    def __init__(self, model: nn.Module, params_arena: torch.Tensor,
                 grads_arena: torch.Tensor, n_live: int, spec: ShardSpec,
                 comm: CommBackend, rank: int, opt: ShardedAdamW,
                 overlap: bool = True):
        assert params_arena.numel() == spec.total, \
            "params arena must be padded to spec.total for in-place all-gather"
        assert grads_arena.numel() == spec.total
        self.model = model
        self.params = params_arena
        self.grads = grads_arena
        self.n = n_live
        self.spec = spec
        self.comm = comm
        self.rank = rank
        self.opt = opt
        self.overlap = overlap and comm.world > 1

        self.sync_enabled = False
        self._works: List[Optional[object]] = [None] * spec.nb
        self._launched = [False] * spec.nb
        self._lock = threading.Lock()   # hooks run on autograd worker threads

        # ---- producer enumeration (coverage accounting)
        notify_ranges, covered = fuse_mod.notify_producers(model)
        hook_params: List[nn.Parameter] = []
        ranges = list(notify_ranges)
        emb_weights = {id(m.weight) for m in model.modules()
                       if isinstance(m, nn.Embedding)}
        for p in model.parameters():
            # AccumulateGrad fires for params not consumed by an arena
            # wrapper, and ALSO for wrapped params used directly as a leaf
            # elsewhere (tied embedding/lm_head weight).
            if id(p) not in covered or id(p) in emb_weights:
                hook_params.append(p)
                ranges.append((p.data.storage_offset(), p.numel()))

        be, nb = spec.bucket_elems, spec.nb
        self._req = [0] * nb
        for off, num in ranges:
            b_lo, b_hi = off // be, (off + num - 1) // be
            for b in range(b_lo, min(b_hi, nb - 1) + 1):
                self._req[b] += min(off + num, (b + 1) * be) - max(off, b * be)
        self._got = [0] * nb

        if self.overlap:
            for p in hook_params:
                off, num = p.data.storage_offset(), p.numel()
                p.register_post_accumulate_grad_hook(self._make_hook(off, num))

    def _make_hook(self, off: int, num: int):
        def hook(_param):
            if self.sync_enabled:
                self._notify(off, num)
        return hook

    def _notify(self, off: int, num: int) -> None:
        """A producer's grads for arena range [off, off+num) have landed."""
        be, nb = self.spec.bucket_elems, self.spec.nb
        b_lo, b_hi = off // be, (off + num - 1) // be
        with self._lock:
            for b in range(b_lo, min(b_hi, nb - 1) + 1):
                self._got[b] += min(off + num, (b + 1) * be) - max(off, b * be)
                if self._got[b] >= self._req[b] and not self._launched[b]:
                    self._launched[b] = True
                    self._works[b] = self.comm.reduce_scatter_bucket_async(
                        self.grads, self.spec, b, self.rank)

    def begin_sync_microbatch(self) -> None:
        """Arm the producers for the last micro-batch of the accumulation
        window."""
        for b in range(self.spec.nb):
            self._got[b] = 0
            self._launched[b] = False
            self._works[b] = None
        self.sync_enabled = True
        if self.overlap:
            fuse_mod.set_grad_notifier(self._notify)

    def finish_step(self, grad_scale: Union[float, torch.Tensor],
                    lr: Optional[float] = None) -> None:
        """After backward: drain per-bucket reduce-scatters (launching any
        not yet launched), fused-AdamW each own segment, all-gather params."""
        self.sync_enabled = False
        if self.overlap:
            fuse_mod.set_grad_notifier(None)
        ag = []
        for b in range(self.spec.nb):
            if self.overlap:
                if not self._launched[b]:   # params with no grad this step
                    self._works[b] = self.comm.reduce_scatter_bucket_async(
                        self.grads, self.spec, b, self.rank)
                self._works[b].wait()
            else:
                self.comm.reduce_scatter_bucket_async(
                    self.grads, self.spec, b, self.rank).wait()
            # AdamW on own segment of the *grad* arena, emitting bf16 params
            # into the *params* arena segment (then gathered in place).
            gseg = self.spec.seg_view(self.grads, b, self.rank)
            pseg = self.spec.seg_view(self.params, b, self.rank)
            from acco_amd import ops
            ops.fused_adamw_step(
                p=self.spec.owned_view(self.opt.p, b), g=gseg,
                m=self.spec.owned_view(self.opt.m, b),
                v=self.spec.owned_view(self.opt.v, b),
                step=self.opt.step_count,
                lr=self.opt.lr if lr is None else lr,
                beta1=self.opt.beta1, beta2=self.opt.beta2, eps=self.opt.eps,
                weight_decay=self.opt.weight_decay, grad_scale=grad_scale,
                out_bf16=pseg, commit=True)
            ag.append(self.comm.all_gather_bucket_async(self.params, self.spec,
                                                        b, self.rank))
        for w in ag:
            w.wait()
        self.opt.step_count += 1

    def zero_grad(self) -> None:
        self.grads.zero_()
