"""LR schedules (reference optim/lr_schedulers.py:16-23 parity:
LinearAnnealingWithWarmUp, plus cosine)."""

from __future__ import annotations

import math


class LinearAnnealingWithWarmup:
    """Linear warmup to ``max_lr`` over ``warmup_steps``, then linear decay
    to ``min_lr`` at ``total_steps``."""

    def __init__(self, optimizer, max_lr: float, warmup_steps: int,
                 total_steps: int, min_lr: float = 0.0):
        self.opt = optimizer
        self.max_lr = max_lr
        self.min_lr = min_lr
        self.warmup_steps = max(warmup_steps, 1)
        self.total_steps = max(total_steps, 1)
        self._step = 0
        self.step()

    def get_lr(self) -> float:
        s = self._step
        if s < self.warmup_steps:
            return self.max_lr * s / self.warmup_steps
        frac = min(1.0, (s - self.warmup_steps) / max(1, self.total_steps - self.warmup_steps))
        return self.max_lr + (self.min_lr - self.max_lr) * frac

    def step(self):
        self._step += 1
        self.opt.set_lr(self.get_lr())

    def state_dict(self):
        return {"step": self._step}

    def load_state_dict(self, sd):
        self._step = sd["step"]
        self.opt.set_lr(self.get_lr())


class CosineAnnealingWithWarmup(LinearAnnealingWithWarmup):
    def get_lr(self) -> float:
        s = self._step
        if s < self.warmup_steps:
            return self.max_lr * s / self.warmup_steps
        frac = min(1.0, (s - self.warmup_steps) / max(1, self.total_steps - self.warmup_steps))
        return self.min_lr + 0.5 * (self.max_lr - self.min_lr) * (1 + math.cos(math.pi * frac))


def build_scheduler(name: str, optimizer, **kw):
    name = (name or "linear").lower()
    if name in ("linearannealingwithwarmup", "linear"):
        return LinearAnnealingWithWarmup(optimizer, **kw)
    if name in ("cosineannealing", "cosine"):
        return CosineAnnealingWithWarmup(optimizer, **kw)
    raise ValueError(f"unknown scheduler {name}")
