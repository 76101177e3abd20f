"""LR schedulers.

InverseSquareRootScheduler parity: reference scheduler.py:8-27. The cosine /
linear warmup schedules the trainers actually use (tiger_trainer.py:223,
rqvae_trainer.py:167 via HF transformers) are provided here directly so the
framework has no hard dependency on transformers for the non-LLM models.
"""

from __future__ import annotations

import math

from torch.optim import Optimizer
from torch.optim.lr_scheduler import LambdaLR


class InverseSquareRootScheduler(LambdaLR):
    def __init__(self, optimizer: Optimizer, warmup_steps: int,
                 last_epoch: int = -1):
        self.warmup_steps = max(1, warmup_steps)

        def fn(step: int) -> float:
            if step < self.warmup_steps:
                return step / self.warmup_steps
            return (self.warmup_steps / max(1, step)) ** 0.5

        super().__init__(optimizer, fn, last_epoch)


def get_linear_schedule_with_warmup(optimizer: Optimizer, num_warmup_steps: int,
                                    num_training_steps: int, last_epoch: int = -1):
    def fn(step: int) -> float:
        if step < num_warmup_steps:
            return step / max(1, num_warmup_steps)
        return max(0.0, (num_training_steps - step)
                   / max(1, num_training_steps - num_warmup_steps))

    return LambdaLR(optimizer, fn, last_epoch)


def get_cosine_schedule_with_warmup(optimizer: Optimizer, num_warmup_steps: int,
                                    num_training_steps: int,
                                    num_cycles: float = 0.5, last_epoch: int = -1):
    def fn(step: int) -> float:
        if step < num_warmup_steps:
            return step / max(1, num_warmup_steps)
        progress = (step - num_warmup_steps) / max(
            1, num_training_steps - num_warmup_steps)
        return max(0.0, 0.5 * (1.0 + math.cos(math.pi * num_cycles * 2.0 * progress)))

    return LambdaLR(optimizer, fn, last_epoch)
