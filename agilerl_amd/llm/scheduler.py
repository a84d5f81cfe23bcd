"""LR schedulers for LLM fine-tuning.

Reference parity: ``agilerl/utils/algo_utils.py``
(create_warmup_cosine_scheduler, consumed at base.py ``update_lr`` :3758).
"""

from __future__ import annotations

import math

import torch

__all__ = ["create_warmup_cosine_scheduler", "WarmupCosineLR"]


class WarmupCosineLR(torch.optim.lr_scheduler.LambdaLR):
    def __init__(
        self,
        optimizer,
        warmup_steps: int,
        total_steps: int,
        min_lr_ratio: float = 0.1,
        last_epoch: int = -1,
    ):
        self.warmup_steps = max(int(warmup_steps), 1)
        self.total_steps = max(int(total_steps), self.warmup_steps + 1)
        self.min_lr_ratio = float(min_lr_ratio)

        def fn(step: int) -> float:
            if step < self.warmup_steps:
                return step / self.warmup_steps
            progress = (step - self.warmup_steps) / (self.total_steps - self.warmup_steps)
            progress = min(progress, 1.0)
            cos = 0.5 * (1.0 + math.cos(math.pi * progress))
            return self.min_lr_ratio + (1 - self.min_lr_ratio) * cos

        super().__init__(optimizer, fn, last_epoch)


def create_warmup_cosine_scheduler(
    optimizer, total_steps: int, warmup_ratio: float = 0.03, min_lr_ratio: float = 0.1
) -> WarmupCosineLR:
    inner = getattr(optimizer, "optimizer", optimizer)  # unwrap OptimizerWrapper
    return WarmupCosineLR(
        inner, warmup_steps=int(total_steps * warmup_ratio), total_steps=total_steps,
        min_lr_ratio=min_lr_ratio,
    )
