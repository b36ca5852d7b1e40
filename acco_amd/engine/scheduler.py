"""LR schedules with batch-advance.

HF ``get_scheduler``-equivalent shapes (the reference uses
``get_scheduler('cosine', num_warmup_steps=warmup,
num_training_steps=nb_grad_tot)`` and advances the counter by the globally
summed grad count per com round — trainer_decoupled.py:102-104,310-315).
``advance(n)`` moves the schedule forward n grad-steps in one call.
"""

from __future__ import annotations

import math


class LRSchedule:
    def __init__(self, base_lr: float, num_warmup_steps: int,
                 num_training_steps: int, kind: str = "cosine"):
        if kind not in ("cosine", "linear", "constant"):
            raise ValueError(f"unknown scheduler kind {kind!r}")
        self.base_lr = float(base_lr)
        self.warmup = int(num_warmup_steps)
        self.total = int(num_training_steps)
        self.kind = kind
        self.current_step = 0

    def factor(self, step: int | None = None) -> float:
        s = self.current_step if step is None else step
        if self.warmup > 0 and s < self.warmup:
            return s / max(1, self.warmup)
        if self.kind == "constant":
            return 1.0
        progress = (s - self.warmup) / max(1, self.total - self.warmup)
        progress = min(max(progress, 0.0), 1.0)
        if self.kind == "linear":
            return 1.0 - progress
        return 0.5 * (1.0 + math.cos(math.pi * progress))

    def lr(self) -> float:
        return self.base_lr * self.factor()

    def advance(self, n: int = 1) -> float:
        self.current_step += int(n)
        return self.lr()

    def state_dict(self) -> dict:
        return {"current_step": self.current_step}

    def load_state_dict(self, sd: dict) -> None:
        self.current_step = int(sd["current_step"])
