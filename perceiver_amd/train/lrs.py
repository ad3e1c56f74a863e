"""LR schedulers (parity with /root/reference/perceiver/scripts/lrs.py:7-38)."""
from __future__ import annotations

import math

from torch.optim.lr_scheduler import LambdaLR


class CosineWithWarmupLR(LambdaLR):
    """Linear warmup then cosine decay to ``min_fraction`` of the peak LR over
    ``training_steps`` steps."""

    def __init__(self, optimizer, training_steps: int, warmup_steps: int,
                 min_fraction: float = 0.0, last_epoch: int = -1):
        self.training_steps = training_steps
        self.warmup_steps = warmup_steps
        self.min_fraction = min_fraction

        def lr_lambda(step: int) -> float:
            if step < warmup_steps:
                return step / max(1, warmup_steps)
            progress = (step - warmup_steps) / max(1, self.training_steps - warmup_steps)
            progress = min(progress, 1.0)
            cos = 0.5 * (1.0 + math.cos(math.pi * progress))
            return min_fraction + (1.0 - min_fraction) * cos

        super().__init__(optimizer, lr_lambda, last_epoch)


class ConstantWithWarmupLR(LambdaLR):
    """Linear warmup then constant LR."""

    def __init__(self, optimizer, warmup_steps: int, last_epoch: int = -1):
        self.warmup_steps = warmup_steps

        def lr_lambda(step: int) -> float:
            if step < warmup_steps:
                return step / max(1, warmup_steps)
            return 1.0

        super().__init__(optimizer, lr_lambda, last_epoch)
