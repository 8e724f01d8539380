"""LR schedulers: cosine / linear / constant with linear warmup
(reference trainer.py:3438-3530 _setup_scheduler)."""

from __future__ import annotations

import math
from typing import Optional


class WarmupScheduler:
    """Step-based scheduler driving FlatAdamW (or any optimizer exposing
    `groups` with a `lr` attribute)."""

    def __init__(self, optimizer, total_steps: int, warmup_steps: int,
                 kind: str = "cosine", min_lr: float = 1e-6):
        assert kind in ("cosine", "linear", "constant")
        self.optimizer = optimizer
        self.total_steps = max(1, total_steps)
        self.warmup_steps = max(0, warmup_steps)
        self.kind = kind
        self.min_lr = min_lr
        self.base_lrs = [g.lr for g in optimizer.groups]
        self.step_num = 0
        self._apply()

    def multiplier(self, step: int) -> float:
        if self.warmup_steps > 0 and step < self.warmup_steps:
            return (step + 1) / self.warmup_steps
        if self.kind == "constant":
            return 1.0
        span = max(1, self.total_steps - self.warmup_steps)
        t = min(1.0, (step - self.warmup_steps) / span)
        if self.kind == "linear":
            return max(0.0, 1.0 - t)
        return 0.5 * (1.0 + math.cos(math.pi * t))

    def _apply(self):
        m = self.multiplier(self.step_num)
        for g, base in zip(self.optimizer.groups, self.base_lrs):
            g.lr = max(self.min_lr, base * m)

    def step(self):
        self.step_num += 1
        self._apply()

    def rebind(self, optimizer):
        """Point the schedule at a REBUILT optimizer (expert add/prune under
        ZeRO-3 replaces the optimizer object): keep the step position, take
        the new group structure, re-apply the current multiplier."""
        base = self.base_lrs[0] if self.base_lrs else None
        self.optimizer = optimizer
        self.base_lrs = [base if base is not None else g.lr
                         for g in optimizer.groups]
        self._apply()

    def get_last_lr(self):
        return [g.lr for g in self.optimizer.groups]

    def set_base_lr(self, lr: float):
        """Adaptive-LR override support: rebase the schedule on a new LR.
        Does NOT immediately re-apply — the caller holds the override LR until
        its grace period expires (trainer suppresses step() meanwhile)."""
        self.base_lrs = [lr for _ in self.base_lrs]

    def state_dict(self):
        return {"step_num": self.step_num, "base_lrs": self.base_lrs,
                "kind": self.kind, "total_steps": self.total_steps,
                "warmup_steps": self.warmup_steps, "min_lr": self.min_lr}

    def load_state_dict(self, sd):
        self.step_num = sd["step_num"]
        self.base_lrs = sd["base_lrs"]
        self.kind = sd.get("kind", self.kind)
        self.total_steps = sd.get("total_steps", self.total_steps)
        self.warmup_steps = sd.get("warmup_steps", self.warmup_steps)
        self.min_lr = sd.get("min_lr", self.min_lr)
        self._apply()


def create_scheduler(optimizer, config, total_steps: int) -> Optional[WarmupScheduler]:
    if not getattr(config, "use_lr_scheduler", True):
        return None
    warmup = int(total_steps * config.warmup_ratio)
    return WarmupScheduler(optimizer, total_steps, warmup,
                           kind=config.lr_scheduler, min_lr=config.min_lr)
