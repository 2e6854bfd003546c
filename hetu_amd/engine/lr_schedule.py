"""Learning-rate schedules as multipliers on the optimizer's base lr.

The reference keeps lr static in the C++ update op and lets the trainer
scripts scale it; here schedules are plain callables `step -> multiplier`
that stay valid UNDER hipGraph capture: the multiplier folds into the
pinned bias-correction scalar the captured Adam kernel re-reads each
replay (`AdamStepOp._lr_scale`), so the recorded graph never changes.
"""
from __future__ import annotations

import math
from typing import Callable

Schedule = Callable[[int], float]


def constant() -> Schedule:
    return lambda step: 1.0


def linear_warmup(warmup_steps: int) -> Schedule:
    def f(step: int) -> float:
        return min(1.0, (step + 1) / max(1, warmup_steps))
    return f


def cosine_with_warmup(warmup_steps: int, total_steps: int,
                       min_ratio: float = 0.1) -> Schedule:
    """Linear warmup then cosine decay to min_ratio * base_lr (the
    standard GPT pretraining schedule)."""
    def f(step: int) -> float:
        if step < warmup_steps:
            return (step + 1) / max(1, warmup_steps)
        t = (step - warmup_steps) / max(1, total_steps - warmup_steps)
        t = min(1.0, t)
        return min_ratio + (1 - min_ratio) * 0.5 * (1 + math.cos(
            math.pi * t))
    return f


def inverse_sqrt(warmup_steps: int) -> Schedule:
    def f(step: int) -> float:
        s = step + 1
        if s < warmup_steps:
            return s / max(1, warmup_steps)
        return math.sqrt(warmup_steps / s)
    return f
