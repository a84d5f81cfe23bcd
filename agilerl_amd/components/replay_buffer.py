"""Experience replay buffers.

Reference parity: ``agilerl/components/replay_buffer.py`` — ReplayBuffer
:29 (circular, batched vector-env adds :119), MultiStepReplayBuffer :219,
PrioritizedReplayBuffer :343 (proportional PER).

MI355X-native design notes:

- Storage is pre-allocated flat torch tensors.  ``storage_device`` may be
  ``"cuda"`` (replay lives in HBM — 288 GB fits tens of millions of
  transitions) or ``"cpu"`` with pinned pages so sampled batches stream to
  HBM with ``non_blocking=True`` copies on a side stream.
- n-step returns are computed **at sample time** from stored windows
  (``ops.nstep_scan`` — HIP kernel on GPU) instead of the reference's
  per-env CPU deques (``replay_buffer.py:287``).
- PER trees are device-resident (see ``segment_tree.py``): priority
  updates and the sampling descent never touch the host.
"""

from __future__ import annotations

from typing import Any, Dict, Optional, Tuple

import numpy as np
import torch

from .. import ops
from .data import to_tensor, tree_index, tree_map
from .segment_tree import MinSegmentTree, SumSegmentTree

__all__ = ["ReplayBuffer", "MultiStepReplayBuffer", "PrioritizedReplayBuffer"]


def _alloc_like(example: torch.Tensor, max_size: int, device: str, pin: bool) -> torch.Tensor:
    t = torch.empty((max_size, *example.shape), dtype=example.dtype, device=device)
    if pin and device == "cpu" and torch.cuda.is_available():
        t = t.pin_memory()
    return t


class ReplayBuffer:
    """Uniform circular replay buffer (batched vector-env transitions)."""

    def __init__(
        self,
        max_size: int,
        device: str = "cpu",
        storage_device: Optional[str] = None,
        pin_memory: bool = True,
        seed: Optional[int] = None,
        dtype: Optional[torch.dtype] = None,
    ):
        self.max_size = int(max_size)
        self.device = device
        self.storage_device = storage_device or "cpu"
        self.pin_memory = pin_memory
        # reference components/replay_buffer.py `dtype`: storage dtype for
        # float data (e.g. torch.float16 halves resident bytes); learners
        # upcast on consumption
        self.dtype = dtype
        self._storage: Optional[Dict[str, Any]] = None
        self._ptr = 0
        self._size = 0
        self.num_envs = 1
        # unseeded buffers derive their stream from the global numpy RNG so
        # np.random.seed(...) upstream makes the whole run reproducible
        if seed is None:
            seed = int(np.random.randint(0, 2**31 - 1))
        self._rng = np.random.default_rng(seed)

    def __len__(self) -> int:
        return self._size

    @property
    def size(self) -> int:
        return self._size

    # ------------------------------------------------------------------
    def _init_storage(self, sample: Dict[str, Any]) -> None:
        self._storage = tree_map(
            lambda t: _alloc_like(t, self.max_size, self.storage_device, self.pin_memory),
            sample,
        )

    def _coerce(self, data: Dict[str, Any]) -> Tuple[Dict[str, Any], int]:
        """Convert to tensors, normalize to a leading batch dim."""
        data = {k: to_tensor(v) for k, v in data.items() if v is not None}
        ref = data.get("reward", next(iter(data.values())))
        while isinstance(ref, dict):
            ref = next(iter(ref.values()))
        if ref.dim() == 0:
            data = tree_map(lambda t: t.unsqueeze(0), data)
            batch = 1
        else:
            batch = ref.shape[0]
        # float64 -> float32 (envs often emit float64 rewards)
        data = tree_map(lambda t: t.float() if t.dtype == torch.float64 else t, data)
        if self.dtype is not None:
            data = tree_map(
                lambda t: t.to(self.dtype) if t.is_floating_point() else t, data
            )
        return data, batch

    @torch.no_grad()
    def add(self, *args, **kwargs) -> None:
        """add(obs=..., action=..., reward=..., next_obs=..., done=...)

        Accepts env-batched arrays (N, ...) or single transitions.
        """
        if args and not kwargs:
            (data,) = args
            if hasattr(data, "to_dict"):
                data = data.to_dict()
        else:
            data = kwargs
        for key in ("obs", "next_obs"):
            if isinstance(data.get(key), (tuple, list)):
                data[key] = {str(i): v for i, v in enumerate(data[key])}
        data, batch = self._coerce(data)
        if self._storage is None:
            self.num_envs = batch
            self._init_storage(tree_index(data, 0))
        idx = (self._ptr + torch.arange(batch)) % self.max_size
        for key, val in data.items():
            tree_map_write(self._storage[key], val, idx)
        self._ptr = int((self._ptr + batch) % self.max_size)
        self._size = min(self._size + batch, self.max_size)

    # ------------------------------------------------------------------
    def _gather(self, idx: torch.Tensor) -> Dict[str, Any]:
        out = tree_map(lambda t: t[idx], self._storage)
        return tree_map(lambda t: t.to(self.device, non_blocking=True), out)

    def sample(self, batch_size: int, return_idx: bool = False):
        idx = torch.from_numpy(self._rng.integers(0, self._size, size=batch_size))
        batch = self._gather(idx)
        return (batch, idx) if return_idx else batch

    def clear(self) -> None:
        self._storage = None
        self._ptr = 0
        self._size = 0


def tree_map_write(store, val, idx):
    if isinstance(store, dict):
        for k in store:
            tree_map_write(store[k], val[k], idx)
    else:
        store[idx] = val.to(store.device, store.dtype)


class MultiStepReplayBuffer(ReplayBuffer):
    """n-step replay: returns computed at sample time from stored windows."""

    def __init__(
        self,
        max_size: int,
        n_step: int = 3,
        gamma: float = 0.99,
        device: str = "cpu",
        storage_device: Optional[str] = None,
        pin_memory: bool = True,
        seed: Optional[int] = None,
    ):
        super().__init__(max_size, device, storage_device, pin_memory, seed=seed)
        self.n_step = int(n_step)
        self.gamma = float(gamma)

    def _valid_span(self) -> int:
        """Number of sampleable entries (oldest-first age order)."""
        return self._size - (self.n_step - 1) * self.num_envs

    def sample(self, batch_size: int, return_idx: bool = False,
               include_one_step: bool = False):
        span = self._valid_span()
        if span <= 0:
            raise RuntimeError("MultiStepReplayBuffer: not enough data for n-step windows")
        age = torch.from_numpy(self._rng.integers(0, span, size=batch_size))
        oldest = self._ptr if self._size == self.max_size else 0
        flat = (oldest + age) % self.max_size
        N = self.num_envs
        # (B, n) same-env windows
        window = (flat.unsqueeze(1) + torch.arange(self.n_step).unsqueeze(0) * N) % self.max_size

        sd = self._storage
        rewards_w = sd["reward"][window].to(self.device)  # (B, n)
        dones_w = sd["done"][window].float().to(self.device)
        returns, steps = ops.nstep_scan(rewards_w, dones_w, self.gamma)
        last = window.gather(1, (steps.long().clamp(min=1) - 1).cpu().unsqueeze(1)).squeeze(1)

        batch = {
            "obs": tree_map(lambda t: t[flat].to(self.device, non_blocking=True), sd["obs"]),
            "action": sd["action"][flat].to(self.device, non_blocking=True),
            "reward": returns.to(self.device),
            "next_obs": tree_map(lambda t: t[last].to(self.device, non_blocking=True), sd["next_obs"]),
            "done": sd["done"][last].to(self.device, non_blocking=True),
            "n_steps": steps.to(self.device),
        }
        if include_one_step:
            # 1-step view of the same anchors (reference dqn_rainbow.py:124
            # combined_reward: sum of 1-step and n-step losses)
            batch["reward_1step"] = sd["reward"][flat].to(self.device, non_blocking=True)
            batch["next_obs_1step"] = tree_map(
                lambda t: t[flat].to(self.device, non_blocking=True), sd["next_obs"]
            )
            batch["done_1step"] = sd["done"][flat].to(self.device, non_blocking=True)
        return (batch, flat) if return_idx else batch


class PrioritizedReplayBuffer(ReplayBuffer):
    """Proportional PER with device-resident sum/min trees."""

    def __init__(
        self,
        max_size: int,
        alpha: float = 0.6,
        device: str = "cpu",
        storage_device: Optional[str] = None,
        pin_memory: bool = True,
        n_step: int = 1,
        gamma: float = 0.99,
        seed: Optional[int] = None,
    ):
        super().__init__(max_size, device, storage_device, pin_memory, seed=seed)
        self.alpha = float(alpha)
        self.n_step = int(n_step)
        self.gamma = float(gamma)
        tree_device = device if str(device).startswith("cuda") else "cpu"
        self.sum_tree = SumSegmentTree(self.max_size, device=tree_device)
        self.min_tree = MinSegmentTree(self.max_size, device=tree_device)
        self.max_priority = 1.0

    @torch.no_grad()
    def add(self, *args, **kwargs) -> None:
        start = self._ptr
        super().add(*args, **kwargs)
        end = self._ptr if self._ptr > start else self._ptr + self.max_size
        idx = torch.arange(start, end) % self.max_size
        prio = torch.full((idx.numel(),), self.max_priority**self.alpha)
        self.sum_tree.update(idx, prio)
        self.min_tree.update(idx, prio)

    def sample(self, batch_size: int, beta: float = 0.4,
               include_one_step: bool = False):
        from ..ops.backend import extension, use_hip

        device = self.sum_tree.device
        ext = extension()
        if use_hip(self.sum_tree.tree) and ext is not None and hasattr(ext, "per_sample"):
            # fused HIP path: stratified prefixes + LDS descent + IS weights
            # in ONE launch; total and p_min are read from the tree roots on
            # device (zero host syncs in the priority-sample math)
            rand01 = torch.rand(batch_size, device=device)
            idx, weights = ext.per_sample(
                self.sum_tree.tree, self.min_tree.tree, rand01,
                int(self._size), float(beta),
            )
        else:
            total = self.sum_tree.sum(0, self._size)
            # stratified prefix sampling
            bounds = torch.linspace(0, total, batch_size + 1, device=device)
            u = bounds[:-1] + torch.rand(batch_size, device=device) * (bounds[1:] - bounds[:-1])
            idx = self.sum_tree.retrieve(u).clamp_(max=self._size - 1)

            p = self.sum_tree.get(idx) / max(total, 1e-12)
            p_min = self.min_tree.min() / max(total, 1e-12)
            max_weight = (p_min * self._size) ** (-beta) if p_min > 0 else 1.0
            weights = ((p * self._size).clamp(min=1e-12) ** (-beta)) / max_weight

        if self.n_step > 1:
            batch = self._nstep_gather(idx.cpu(), include_one_step=include_one_step)
        else:
            batch = self._gather(idx.cpu())
        batch["weights"] = weights.to(self.device)
        batch["idxs"] = idx.to(self.device)
        return batch

    def _nstep_gather(self, idx: torch.Tensor,
                      include_one_step: bool = False) -> Dict[str, Any]:
        """n-step windows anchored at ``idx`` (clamped away from the write head)."""
        N = self.num_envs
        oldest = self._ptr if self._size == self.max_size else 0
        span = max(self._size - (self.n_step - 1) * N, 1)
        age = (idx - oldest) % self.max_size
        age = age.clamp(max=span - 1)
        flat = (oldest + age) % self.max_size
        window = (flat.unsqueeze(1) + torch.arange(self.n_step).unsqueeze(0) * N) % self.max_size
        sd = self._storage
        rewards_w = sd["reward"][window].to(self.device)
        dones_w = sd["done"][window].float().to(self.device)
        returns, steps = ops.nstep_scan(rewards_w, dones_w, self.gamma)
        last = window.gather(1, (steps.long().clamp(min=1) - 1).cpu().unsqueeze(1)).squeeze(1)
        out = {
            "obs": tree_map(lambda t: t[flat].to(self.device, non_blocking=True), sd["obs"]),
            "action": sd["action"][flat].to(self.device, non_blocking=True),
            "reward": returns.to(self.device),
            "next_obs": tree_map(lambda t: t[last].to(self.device, non_blocking=True), sd["next_obs"]),
            "done": sd["done"][last].to(self.device, non_blocking=True),
            "n_steps": steps.to(self.device),
        }
        if include_one_step:
            out["reward_1step"] = sd["reward"][flat].to(self.device, non_blocking=True)
            out["next_obs_1step"] = tree_map(
                lambda t: t[flat].to(self.device, non_blocking=True), sd["next_obs"]
            )
            out["done_1step"] = sd["done"][flat].to(self.device, non_blocking=True)
        return out

    @torch.no_grad()
    def update_priorities(self, idx: torch.Tensor, priorities: torch.Tensor) -> None:
        priorities = priorities.detach().reshape(-1).clamp(min=1e-8)
        prio_a = priorities**self.alpha
        idx = idx.reshape(-1)
        self.sum_tree.update(idx, prio_a)
        self.min_tree.update(idx, prio_a)
        self.max_priority = max(self.max_priority, float(priorities.max()))
