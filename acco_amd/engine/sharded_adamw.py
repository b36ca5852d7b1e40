"""ZeRO-1 sharded AdamW on the bucket-major flat buffer.

Each rank owns nb contiguous (bucket, rank) segments of the flat com buffer
(parallel/comm.py ShardSpec); the fp32 master params + AdamW state live in
one contiguous bucket-major vector of spec.owned elements. `step_bucket`
runs the fused gfx950 AdamW kernel (ops.fused_adamw_step) on one bucket:
it reads the freshly reduce-scattered bf16 gradient segment *in place* in
the com buffer, does fp32 AdamW math, and overwrites the same segment with
the updated bf16 parameters ready for the all-gather — K3+K4+K5+K7 of
SURVEY.md §2.5 in one pass over memory.

The ACCO tentative step (even com rounds) is `commit=False`: updated bf16
params are emitted but p/m/v/step are untouched — algebraically identical
to the reference's snapshot → step → rollback
(trainer_decoupled.py:79-84,113-125) at zero state-copy cost.
"""

from __future__ import annotations

from typing import Optional, Union

import torch

from acco_amd import ops
from acco_amd.parallel.comm import ShardSpec


class ShardedAdamW:
    def __init__(self, spec: ShardSpec, rank: int, device: torch.device,
                 lr: float, betas=(0.9, 0.95), eps: float = 1e-8,
                 weight_decay: float = 0.1):
        self.spec = spec
        self.rank = rank
        self.lr = float(lr)
        self.beta1, self.beta2 = float(betas[0]), float(betas[1])
        self.eps = float(eps)
        self.weight_decay = float(weight_decay)
        self.step_count = 0
        self.p = torch.zeros(spec.owned, dtype=torch.float32, device=device)
        self.m = torch.zeros_like(self.p)
        self.v = torch.zeros_like(self.p)

    @torch.no_grad()
    def init_master_from_buffer(self, buffer: torch.Tensor) -> None:
        """Seed the fp32 master shard from the (bf16) param values currently
        in the buffer's own segments (reference trainer_decoupled.py:297-299)."""
        for j in range(self.spec.nb):
            self.spec.owned_view(self.p, j).copy_(
                self.spec.seg_view(buffer, j, self.rank))

    @torch.no_grad()
    def step_bucket(self, j: int, buffer: torch.Tensor,
                    grad_scale: Union[float, torch.Tensor],
                    commit: bool, lr: Optional[float] = None) -> None:
        seg = self.spec.seg_view(buffer, j, self.rank)
        ops.fused_adamw_step(
            p=self.spec.owned_view(self.p, j),
            g=seg,
            m=self.spec.owned_view(self.m, j),
            v=self.spec.owned_view(self.v, j),
            step=self.step_count,
            lr=self.lr if lr is None else lr,
            beta1=self.beta1, beta2=self.beta2, eps=self.eps,
            weight_decay=self.weight_decay,
            grad_scale=grad_scale,
            out_bf16=seg,
            commit=commit,
        )

    def finish_round(self, commit: bool) -> None:
        if commit:
            self.step_count += 1

    @torch.no_grad()
    def consolidate(self, world: int) -> Optional[dict]:
        """Gather the sharded fp32 state into full flat vectors on rank 0
        (world-size-independent checkpoints — the capability of torch's
        ZeroRedundancyOptimizer.consolidate_state_dict which the reference
        relies on implicitly). Returns None on non-zero ranks."""
        import torch.distributed as dist
        spec = self.spec
        full = {}
        for name, shard in (("p", self.p), ("m", self.m), ("v", self.v)):
            if world == 1:
                gathered = [shard]
            else:
                gathered = [torch.empty_like(shard) for _ in range(world)]
                dist.all_gather(gathered, shard)
            if self.rank != 0:
                continue
            out = torch.empty(spec.total, dtype=torch.float32,
                              device=shard.device)
            for r in range(world):
                for j in range(spec.nb):
                    spec.seg_view(out, j, r).copy_(
                        spec.owned_view(gathered[r], j))
            full[name] = out
        if self.rank != 0 and world > 1:
            return None
        full["step"] = self.step_count
        return full

    def state_dict(self) -> dict:
        return {"step": self.step_count, "p": self.p, "m": self.m, "v": self.v,
                "lr": self.lr, "beta1": self.beta1, "beta2": self.beta2,
                "eps": self.eps, "weight_decay": self.weight_decay}

    def load_state_dict(self, sd: dict) -> None:
        self.step_count = int(sd["step"])
        self.p.copy_(sd["p"])
        self.m.copy_(sd["m"])
        self.v.copy_(sd["v"])
