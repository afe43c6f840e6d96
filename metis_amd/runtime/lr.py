"""Learning-rate schedules for FusedAdamW (its ``lr`` is a plain float
read at every step, so a schedule just assigns it before the step).

GPT-3 style: linear warmup to ``lr_max`` then cosine decay to
``lr_min`` over ``decay_steps``, constant ``lr_min`` afterwards.
"""

from __future__ import annotations

import math


class WarmupCosineLR:
    def __init__(self, optimizer, lr_max: float, warmup_steps: int,
                 decay_steps: int, lr_min: float = 0.0):
        assert decay_steps >= warmup_steps >= 0
        self.opt = optimizer
        self.lr_max = lr_max
        self.lr_min = lr_min
        self.warmup_steps = warmup_steps
        self.decay_steps = decay_steps

    def lr_at(self, step: int) -> float:
        if self.warmup_steps and step < self.warmup_steps:
            return self.lr_max * (step + 1) / self.warmup_steps
        if step >= self.decay_steps:
            return self.lr_min
        frac = ((step - self.warmup_steps)
                / max(self.decay_steps - self.warmup_steps, 1))
        return self.lr_min + 0.5 * (self.lr_max - self.lr_min) * (
            1.0 + math.cos(math.pi * frac))

    def step(self, step: int) -> float:
        """Set the optimizer lr for ``step`` (0-based) and return it."""
        lr = self.lr_at(step)
        self.opt.lr = lr
        return lr
