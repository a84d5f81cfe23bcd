"""Uniform sampling facade over the buffer family.

Reference parity: ``agilerl/components/sampler.py:21``.
"""

from __future__ import annotations

from typing import Optional

from .replay_buffer import MultiStepReplayBuffer, PrioritizedReplayBuffer, ReplayBuffer

__all__ = ["Sampler"]


class Sampler:
    def __init__(self, memory=None, per: bool = False, n_step: bool = False):
        self.memory = memory
        self.per = per or isinstance(memory, PrioritizedReplayBuffer)
        self.n_step = n_step or isinstance(memory, MultiStepReplayBuffer)

    def sample(self, batch_size: int, beta: Optional[float] = None, return_idx: bool = False):
        if self.per:
            return self.memory.sample(batch_size, beta=beta if beta is not None else 0.4)
        return self.memory.sample(batch_size, return_idx=return_idx)
