"""Dynamic loss scaling for fp16 training (reference
``alpa/model/model_util.py`` DynamicScale; bf16 is the default compute
dtype on MI355X and needs no scaling, but the fp16 path keeps parity).
"""
from __future__ import annotations

from typing import Iterable, Optional

import torch


class DynamicScale:

    def __init__(self, init_scale: float = 2.0 ** 15,
                 growth_factor: float = 2.0, backoff_factor: float = 0.5,
                 growth_interval: int = 2000):
        self.scale = init_scale
        self.growth_factor = growth_factor
        self.backoff_factor = backoff_factor
        self.growth_interval = growth_interval
        self._good_steps = 0

    def scale_loss(self, loss: torch.Tensor) -> torch.Tensor:
        return loss * self.scale

    def found_inf(self, grads: Iterable[torch.Tensor]) -> bool:
        for g in grads:
            if g is None:
                continue
            if not torch.isfinite(g.float().sum()):
                return True
        return False

    def unscale_factor(self) -> float:
        """Multiply into the optimizer grad_scale (fuses the unscale into
        the AdamW kernel, no extra pass)."""
        return 1.0 / self.scale

    def update(self, found_inf: bool) -> bool:
        """Returns True when the step should be SKIPPED."""
        if found_inf:
            self.scale = max(1.0, self.scale * self.backoff_factor)
            self._good_steps = 0
            return True
        self._good_steps += 1
        if self._good_steps >= self.growth_interval:
            self.scale *= self.growth_factor
            self._good_steps = 0
        return False
