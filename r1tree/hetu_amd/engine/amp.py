"""AMP: dynamic loss scaling + finite checks.

Reference parity: hetu/graph/autocast/gradscaler.cc (+ UpdateScale.cu /
CheckFinite.cu kernels): scale the loss-gradient seed, check gradients for
inf/nan before the optimizer step, skip-and-backoff on overflow, grow the
scale every `growth_interval` good steps.  MI355X note: the kernel set is
bf16-first (bf16 needs no scaling); the scaler exists for fp16 runs and
for parity — it plugs into the two-phase (grad-buffer) update paths where
gradients are explicit (PipelineRunner, stage update graphs).
"""
from __future__ import annotations

from typing import Iterable, Optional

import torch


class GradScaler:
    def __init__(self, init_scale: float = 2.0 ** 16,
                 growth_factor: float = 2.0, backoff_factor: float = 0.5,
                 growth_interval: int = 200, enabled: bool = True):
        self.scale = init_scale if enabled else 1.0
        self.growth_factor = growth_factor
        self.backoff_factor = backoff_factor
        self.growth_interval = growth_interval
        self.enabled = enabled
        self._good_steps = 0
        self.skipped = 0

    def check_and_update(self, grads: Iterable[torch.Tensor]) -> bool:
        """Returns True if the step should proceed (all grads finite);
        updates the scale either way."""
        if not self.enabled:
            return True
        finite = all(torch.isfinite(g).all().item() for g in grads
                     if g is not None)
        if finite:
            self._good_steps += 1
            if self._good_steps >= self.growth_interval:
                self.scale *= self.growth_factor
                self._good_steps = 0
            return True
        self.skipped += 1
        self.scale = max(1.0, self.scale * self.backoff_factor)
        self._good_steps = 0
        return False

    def unscale_(self, grads: Iterable[torch.Tensor]):
        if self.enabled and self.scale != 1.0:
            inv = 1.0 / self.scale
            for g in grads:
                if g is not None:
                    g.mul_(inv)
