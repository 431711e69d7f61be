"""Learning-rate schedules (constant / linear / cosine with warmup).

Works with both optimizer APIs in this repo: the ZeRO optimizers read
``optimizer.lr`` each step (their ``param_groups`` are ephemeral
views), torch optimizers read ``param_groups[i]["lr"]`` — the schedule
sets both.  Step-based (one ``step()`` per OPTIMIZER step, i.e. per
accumulation window), matching how the trainers count steps.
"""

from __future__ import annotations

import math

__all__ = ["LRSchedule"]


class LRSchedule:
    KINDS = ("constant", "linear", "cosine")

    def __init__(
        self,
        optimizer,
        base_lr: float,
        total_steps: int,
        warmup_steps: int = 0,
        kind: str = "cosine",
        min_lr: float = 0.0,
    ):
        assert kind in self.KINDS, f"lr_schedule must be one of {self.KINDS}"
        assert warmup_steps >= 0 and total_steps >= 1
        self.optimizer = optimizer
        self.base_lr = float(base_lr)
        self.total_steps = int(total_steps)
        self.warmup_steps = int(warmup_steps)
        self.kind = kind
        self.min_lr = float(min_lr)
        self._step = 0

    # ------------------------------------------------------------------
    def lr_at(self, step: int) -> float:
        if self.warmup_steps and step < self.warmup_steps:
            # linear warmup from 0 (step 0 uses base_lr/warmup, not 0)
            return self.base_lr * (step + 1) / self.warmup_steps
        if self.kind == "constant":
            return self.base_lr
        decay_span = max(self.total_steps - self.warmup_steps, 1)
        t = min(max(step - self.warmup_steps, 0), decay_span) / decay_span
        if self.kind == "linear":
            return self.min_lr + (self.base_lr - self.min_lr) * (1.0 - t)
        # cosine
        return self.min_lr + 0.5 * (self.base_lr - self.min_lr) * (
            1.0 + math.cos(math.pi * t)
        )

    def step(self) -> float:
        """Set the lr for the CURRENT optimizer step, then advance."""
        lr = self.lr_at(self._step)
        self._step += 1
        self._apply(lr)
        return lr

    def _apply(self, lr: float) -> None:
        if hasattr(self.optimizer, "lr"):
            self.optimizer.lr = lr  # ZeRO-1/2 read this directly
        for g in getattr(self.optimizer, "param_groups", []):
            if isinstance(g, dict):
                g["lr"] = lr

    # ------------------------------------------------------------------
    def state_dict(self):
        return {"step": self._step}

    def load_state_dict(self, sd):
        self._step = int(sd["step"])
