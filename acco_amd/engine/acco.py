"""The ACCO decoupled-training engine: the two-round state machine.

Reimplements (from scratch, MI355X-first) the algorithm of reference
trainer_decoupled.py:

- `gradient_step`            (:18-39)   — micro-batch forward/backward into
                                          the flat grad arena,
- `update_buffers_step`      (:43-63)   — params ⇄ com-buffer swap,
- `communication_round`      (:67-126)  — count all-reduce + bucketed
                                          reduce-scatter + sharded AdamW +
                                          bucketed all-gather,
- `com_routine` / `train_acco` (:129-168, :431-520) — the compute∥comm
                                          thread choreography on two HIP
                                          streams,
- `warmup_steps`             (:318-383) — sequential warmup rounds,
- `train_dpu`                (:605-663) — the single-stream DPU baseline.

ACCO semantics (must hold exactly — SURVEY.md §3.1): gradients accumulate
across two com rounds and are zeroed only after even rounds; the even round
takes a *tentative* optimizer step (commit=False — no state mutation,
replacing the reference's snapshot/rollback) to predict the next params so
compute never stalls; the odd round takes the true step on the full
two-half-round gradient; the scheduler and the global grad counter advance
only on odd rounds, by the globally-summed grad count (which is what makes
heterogeneous per-rank accumulation correct).
"""

from __future__ import annotations

import threading
import time
from typing import Callable, Dict, Optional

import torch

from acco_amd.engine.sharded_adamw import ShardedAdamW
from acco_amd.engine.scheduler import LRSchedule
from acco_amd.engine.streams import Event, Stream
from acco_amd.parallel.comm import CommBackend, ShardSpec
from acco_amd.utils.profiling import trace_range


class AccoEngine:
    """Owns the flat buffers and runs ACCO / DPU rounds.

    Parameters
    ----------
    params_arena : flat (bf16) parameter vector, model params alias it.
    grads_arena  : flat grad vector of the same padded size (spec.total).
    spec         : bucket-major shard geometry.
    comm         : collective backend (RCCL / gloo).
    forward_backward : callable(inputs) -> loss float tensor; must
        accumulate gradients into grads_arena (i.e. call loss.backward()).
    """

    def __init__(self, *, params_arena: torch.Tensor, grads_arena: torch.Tensor,
                 n_live: int, spec: ShardSpec, comm: CommBackend, rank: int,
                 device: torch.device, opt: ShardedAdamW, sched: LRSchedule,
                 forward_backward: Callable[[Dict], torch.Tensor],
                 next_batch: Callable[[], Dict],
                 n_grad_accumulation: int = 1,
                 grad_reduce_dtype: Optional[str] = None,
                 log=None):
        self.params = params_arena
        self.grads = grads_arena
        self.n = n_live
        self.spec = spec
        self.comm = comm
        self.rank = rank
        self.device = device
        self.opt = opt
        self.sched = sched
        self.forward_backward = forward_backward
        self.next_batch = next_batch
        self.n_acc = n_grad_accumulation
        self.log = log

        # the communication buffer (reference com_buffer, :244-269)
        self.com_buffer = torch.zeros(spec.total, dtype=params_arena.dtype,
                                      device=device)
        # fp32-accumulate reduce (train.grad_reduce_dtype='fp32'): the com
        # round runs on an fp32 shadow — cast in, reduce-scatter + AdamW +
        # all-gather in fp32, cast back — gating the bf16 8-rank summation
        # error the reference accepts (trainer_decoupled.py:88-93; SURVEY.md
        # §7 "bf16 grad reduce-scatter numerics"). 2× collective bytes.
        self._com32: Optional[torch.Tensor] = None
        if (grad_reduce_dtype == "fp32"
                and self.com_buffer.dtype == torch.bfloat16):
            self._com32 = torch.zeros(spec.total, dtype=torch.float32,
                                      device=device)
        int_t = torch.int32
        self.count_grad_local = torch.zeros(1, dtype=int_t, device=device)
        self.count_grad_this_round = torch.zeros(1, dtype=int_t, device=device)
        self.loss_static = torch.zeros(1, dtype=torch.float32, device=device)

        self.com_stream = Stream(device)
        self.grad_stream = Stream(device)
        self.com_event = Event(device)
        self.update_event = Event(device)
        self.end_of_grad = Event(device)

        self.round_idx = 0            # reference count_after_init
        self.count_grad_tot = 0
        self.com_finished = threading.Event()

        # arena-swap: set by enable_arena_swap (trainer/bench); None keeps
        # the copy path (engine-level tests, DDP mode)
        self.flip_param_views: Optional[Callable[[torch.Tensor], None]] = None
        # hooks the trainer installs (logging / eval / checkpoint); called
        # on the compute thread after each buffer update, rank 0 only.
        self.on_round_complete: Optional[Callable[[int, int], None]] = None
        # test instrumentation: when set to a list, records the per-round
        # LOCAL grad count at each buffer update (oracle replay support)
        self.trace: Optional[list] = None
        # bench instrumentation: called on EVERY rank after each com-round
        # boundary (round_idx already incremented); python-side micro-batch
        # counter for token accounting
        self.on_round_boundary: Optional[Callable[[int], None]] = None
        self.micro_steps = 0
        # com-log: per-round wall time of the communication round, dumped by
        # the trainer at end of training (the reference's save_com_logs
        # channel, utils/logs_utils.py:141-152)
        self.com_log: list = []
        # ACCO_DEBUG_HANDOFF=1: checksum the com-buffer ownership handoff
        # between the comm and compute threads (SURVEY.md §5 race-detection
        # rebuild note)
        import os as _os
        self._debug_handoff = _os.environ.get("ACCO_DEBUG_HANDOFF") == "1"
        self._handoff_sum: Optional[torch.Tensor] = None

    # ------------------------------------------------------------ pieces

    def gradient_step(self, inputs: Dict) -> None:
        """One micro-batch forward/backward accumulating into grads arena
        (reference gradient_step :18-39)."""
        with trace_range("acco/gradient_step"):
            loss = self.forward_backward(inputs)
        self.count_grad_local += 1
        self.micro_steps += 1
        self.loss_static.copy_(loss.detach().float().reshape(1))

    def enable_arena_swap(self, model, grads_arena: torch.Tensor) -> None:
        """Arena-swap optimization: after a com round the com buffer holds
        the NEW parameters, so `update_buffers_step` swaps the params-arena
        and com-buffer ROLES (re-pointing the model's param views — a
        CPU-side metadata walk) instead of copying 2·N bytes on the compute
        stream. Fused projection views are pre-derived for BOTH arenas and
        flipped with the swap."""
        from acco_amd.engine import arena as arena_mod
        from acco_amd.models import fuse as fuse_mod
        snap_a = fuse_mod.snapshot_fused_views(model)
        a, b = self.params, self.com_buffer
        arena_mod.repoint_params(model, b)
        fuse_mod.install_fused_projections(model, b, grads_arena)
        snap_b = fuse_mod.snapshot_fused_views(model)
        arena_mod.repoint_params(model, a)
        fuse_mod.apply_fused_views(snap_a)
        b_ptr = b.data_ptr()

        def flip(active: torch.Tensor) -> None:
            arena_mod.repoint_params(model, active)
            fuse_mod.apply_fused_views(
                snap_b if active.data_ptr() == b_ptr else snap_a)

        self.flip_param_views = flip

    @torch.no_grad()
    def update_buffers_step(self, zero_grads: bool) -> None:
        """Swap: params ← buffer (new params); buffer ← accumulated grads;
        publish local grad count; optionally zero grads+count
        (reference update_buffers_step :43-63). With arena-swap enabled the
        first copy becomes a role swap (zero bytes moved)."""
        if self._debug_handoff and self._handoff_sum is not None:
            now = self.com_buffer[:min(self.n, 1 << 20)].float().sum()
            assert torch.equal(now, self._handoff_sum), (
                "com-buffer ownership handoff violated: buffer changed "
                "between com_finished and update_buffers_step")
        if self.flip_param_views is not None:
            self.params, self.com_buffer = self.com_buffer, self.params
            self.flip_param_views(self.params)
        else:
            self.params[:self.n].copy_(self.com_buffer[:self.n])
        self.com_buffer[:self.n].copy_(self.grads[:self.n])
        if self.com_buffer.numel() > self.n:
            self.com_buffer[self.n:].zero_()
        self.count_grad_this_round.copy_(self.count_grad_local)
        if self.trace is not None:
            self.trace.append(int(self.count_grad_local.item()))
        if zero_grads:
            self.grads.zero_()
            self.count_grad_local.zero_()

    @torch.no_grad()
    def communication_round(self, commit: bool, advance_sched: bool = None) -> int:
        """One com round on the com buffer: C2 count all-reduce (async) →
        per-bucket C3 reduce-scatter → fused sharded AdamW per bucket as its
        bucket lands (pipelined) → per-bucket C4 all-gather
        (reference communication_step :67-126). Returns the global grad
        count of the round."""
        if advance_sched is None:
            advance_sched = commit
        t0 = time.time()
        with trace_range("acco/communication_round"):
            n = self._communication_round(commit, advance_sched)
        self.com_log.append({"round": self.round_idx, "commit": commit,
                             "t": time.time() - t0, "count": n})
        if self._debug_handoff:
            self._handoff_sum = self.com_buffer[:min(self.n, 1 << 20)].float().sum()
        return n

    @torch.no_grad()
    def _communication_round(self, commit: bool, advance_sched: bool) -> int:
        work_count = self.comm.all_reduce_sum_async(self.count_grad_this_round)
        buf = self.com_buffer
        if self._com32 is not None:
            self._com32.copy_(self.com_buffer)     # grads bf16 → fp32
            buf = self._com32
        rs = [self.comm.reduce_scatter_bucket_async(buf, self.spec,
                                                    j, self.rank)
              for j in range(self.spec.nb)]
        work_count.wait()
        # grad averaging divides by the GLOBAL grad count (reference :85-98)
        inv_count = 1.0 / self.count_grad_this_round.float()
        lr = self.sched.lr()
        ag = []
        for j in range(self.spec.nb):
            rs[j].wait()
            self.opt.step_bucket(j, buf, grad_scale=inv_count,
                                 commit=commit, lr=lr)
            ag.append(self.comm.all_gather_bucket_async(buf, self.spec, j,
                                                        self.rank))
        for w in ag:
            w.wait()
        if buf is not self.com_buffer:
            self.com_buffer.copy_(buf)             # new params fp32 → bf16
        self.opt.finish_round(commit)
        n_global = int(self.count_grad_this_round.item())
        if advance_sched:
            self.sched.advance(n_global)
        return n_global

    # ------------------------------------------------------- init / warmup

    def bootstrap(self, n_warmup_steps: int) -> None:
        """Fill the com buffer so the first threaded round has gradients
        (reference prepare_buffer_com :262-267 + warmup_steps :318-383 +
        train_acco init :436-442)."""
        if n_warmup_steps > 0:
            # buffer starts holding the params; warmup rounds run sequentially
            self.com_buffer[:self.n].copy_(self.params[:self.n])
            for _ in range(n_warmup_steps):
                self.params[:self.n].copy_(self.com_buffer[:self.n])
                for _ in range(self.n_acc):
                    self.gradient_step(self.next_batch())
                self.update_buffers_step(zero_grads=True)
                self.communication_round(commit=True)
            # tail: one more grad round so the threaded loop starts with
            # fresh grads in the buffer (reference :358-383)
            self.params[:self.n].copy_(self.com_buffer[:self.n])
            for _ in range(self.n_acc):
                self.gradient_step(self.next_batch())
            self.update_buffers_step(zero_grads=True)
            self.count_grad_tot = self.comm.world * (n_warmup_steps + 1) * self.n_acc
        else:
            # bootstrap gradient: one real forward/backward seeds the buffer
            # (reference prepare_grads + init_count=1, :262-267,441)
            self.gradient_step(self.next_batch())
            self.com_buffer[:self.n].copy_(self.grads[:self.n])
            self.count_grad_this_round.fill_(1)
            # grads stay un-zeroed: round 0's compute accumulates onto them
            # and the tentative round consumes the buffered copy (reference
            # semantics — count_grad_local continues from 1).
            self.count_grad_tot = 0
        self.round_idx = 0

    # --------------------------------------------------------- ACCO loop

    def train_acco(self, nb_grad_tot: int, n_warmup_steps: int = 0,
                   max_rounds: Optional[int] = None) -> None:
        self.bootstrap(n_warmup_steps)
        barrier = threading.Barrier(2)
        stop = threading.Event()

        def keep_going() -> bool:
            if max_rounds is not None and self.round_idx >= max_rounds:
                return False
            return self.count_grad_tot < nb_grad_tot

        def com_routine():
            try:
                self.com_stream.wait_default(self.device)
                with self.com_stream.activate():
                    while keep_going() and not stop.is_set():
                        commit = (self.round_idx % 2 == 1)
                        self.communication_round(commit=commit)
                        self.com_event.record_and_sync(self.com_stream)
                        self.com_finished.set()
                        barrier.wait()
            except threading.BrokenBarrierError:
                pass
            except Exception:
                # a dead com thread must not leave the compute thread parked
                barrier.abort()
                raise

        com_thread = threading.Thread(target=com_routine, daemon=True)
        com_thread.start()

        try:
            self.grad_stream.wait_default(self.device)
            with self.grad_stream.activate():
                while keep_going():
                    for _ in range(self.n_acc):
                        self.gradient_step(self.next_batch())
                    self.end_of_grad.record_and_sync(self.grad_stream)
                    if self.com_finished.is_set():
                        self.com_finished.clear()
                        if self.round_idx % 2 == 1:
                            self.count_grad_tot += int(
                                self.count_grad_this_round.item())
                        self.update_buffers_step(
                            zero_grads=(self.round_idx % 2 == 0))
                        self.update_event.record_and_sync(self.grad_stream)
                        self.round_idx += 1
                        barrier.wait()
                        if self.on_round_boundary is not None:
                            self.on_round_boundary(self.round_idx)
                        if self.on_round_complete is not None and self.rank == 0:
                            self.on_round_complete(self.round_idx,
                                                   self.count_grad_tot)
        finally:
            stop.set()
            # unblock the com thread if it is parked at the barrier
            try:
                barrier.abort()
            except Exception:
                pass
            com_thread.join(timeout=60.0)

    # ---------------------------------------------------------- DPU loop

    def train_dpu(self, nb_grad_tot: int, n_warmup_steps: int = 0,
                  max_rounds: Optional[int] = None) -> None:
        """Delayed-parameter-update baseline: same primitives, sequential,
        stale gradients, full step every round (reference train_dpu
        :605-663 — communication_step there always runs with the default
        count_after_init=-1, i.e. commit+schedule every round, while
        update_buffers_step alternates grad zeroing)."""
        self.bootstrap(n_warmup_steps)
        # the grad-zeroing alternation continues from the warmup rounds'
        # parity (reference initializes its round counter to n_warmup_steps,
        # trainer_decoupled.py:618,659 — with an odd warmup the first
        # threaded round must NOT zero)
        count_com = n_warmup_steps
        max_com = None if max_rounds is None else n_warmup_steps + max_rounds
        while self.count_grad_tot < nb_grad_tot and (
                max_com is None or count_com < max_com):
            for _ in range(self.n_acc):
                self.gradient_step(self.next_batch())
            self.communication_round(commit=True)
            self.count_grad_tot += int(self.count_grad_this_round.item())
            self.update_buffers_step(zero_grads=(count_com % 2 == 0))
            count_com += 1
            if self.on_round_boundary is not None:
                self.on_round_boundary(count_com)
            if self.on_round_complete is not None and self.rank == 0:
                self.on_round_complete(count_com, self.count_grad_tot)
