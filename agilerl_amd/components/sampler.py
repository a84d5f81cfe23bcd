"""Uniform sampling facade over the buffer family.

Reference parity: ``agilerl/components/sampler.py:21`` — samples from a
buffer, or (distributed mode, sampler.py:55-63) from a
dataset+dataloader pair: the dataloader shards batches per rank and the
sampler cycles it, updating the dataset's batch size on the fly.
"""

from __future__ import annotations

from typing import Optional

from .replay_buffer import MultiStepReplayBuffer, PrioritizedReplayBuffer, ReplayBuffer

__all__ = ["Sampler"]


class Sampler:
    def __init__(self, memory=None, per: bool = False, n_step: bool = False,
                 dataset=None, dataloader=None):
        if memory is None and (dataset is None or dataloader is None):
            raise ValueError(
                "Sampler needs either 'memory' or ('dataset' AND 'dataloader')"
            )
        self.memory = memory
        self.dataset = dataset
        self.dataloader = dataloader
        self._data_iter = None
        self.per = per or isinstance(memory, PrioritizedReplayBuffer)
        self.n_step = n_step or isinstance(memory, MultiStepReplayBuffer)

    def _sample_from_loader(self, batch_size: int):
        if self.dataset is not None and hasattr(self.dataset, "batch_size"):
            self.dataset.batch_size = batch_size
        if self._data_iter is None:
            self._data_iter = iter(self.dataloader)
        try:
            return next(self._data_iter)
        except StopIteration:
            self._data_iter = iter(self.dataloader)
            return next(self._data_iter)

    def sample(self, batch_size: int, beta: Optional[float] = None,
               return_idx: bool = False, include_one_step: bool = False):
        if self.memory is None:
            return self._sample_from_loader(batch_size)
        if self.per:
            return self.memory.sample(batch_size,
                                      beta=beta if beta is not None else 0.4,
                                      include_one_step=include_one_step)
        if include_one_step and self.n_step:
            return self.memory.sample(batch_size, include_one_step=True)
        return self.memory.sample(batch_size, return_idx=return_idx)
