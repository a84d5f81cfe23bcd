"""On-policy rollout buffer with device-resident GAE.

Reference parity: ``agilerl/components/rollout_buffer.py`` — RolloutBuffer
:70, ``compute_returns_and_advantages`` :472 (reference does a sequential
numpy reverse scan on CPU), flat minibatches :585, BPTT sequence
minibatches :941.

MI355X design: storage is (T, N, ...) torch tensors allocated once on the
target device; the GAE reverse scan runs as a single HIP kernel
(``ops.gae_scan``, one lane per env column) so rollouts collected on GPU
never round-trip to host.
"""

from __future__ import annotations

from typing import Any, Dict, Iterator, List, Optional, Tuple

import numpy as np
import torch

from .. import ops
from .data import to_tensor, tree_map

__all__ = ["RolloutBuffer"]


class RolloutBuffer:
    def __init__(
        self,
        capacity: int,
        num_envs: int,
        device: str = "cpu",
        gamma: float = 0.99,
        gae_lambda: float = 0.95,
        recurrent: bool = False,
        observation_space=None,
        action_space=None,
        use_gae: bool = True,
        wrap_at_capacity: bool = False,
        max_seq_len: Optional[int] = None,
        bptt_sequence_type: str = "chunked",
        hidden_state_architecture=None,
    ):
        self.capacity = int(capacity)
        self.num_envs = int(num_envs)
        self.device = device
        self.gamma = float(gamma)
        self.gae_lambda = float(gae_lambda)
        self.recurrent = recurrent
        # reference rollout_buffer.py constructor surface: spaces are
        # informational (storage shapes come from the first add), use_gae
        # toggles the lambda-weighted scan vs plain discounted returns,
        # wrap_at_capacity keeps adding as a ring instead of raising,
        # max_seq_len/bptt_sequence_type are the BPTT slicing defaults
        self.observation_space = observation_space
        self.action_space = action_space
        self.use_gae = bool(use_gae)
        self.wrap_at_capacity = bool(wrap_at_capacity)
        self.max_seq_len = max_seq_len
        self.bptt_sequence_type = bptt_sequence_type
        self.hidden_state_architecture = hidden_state_architecture

        self._storage: Optional[Dict[str, Any]] = None
        self.pos = 0
        self.full = False
        self.advantages: Optional[torch.Tensor] = None
        self.returns: Optional[torch.Tensor] = None

    def __len__(self) -> int:
        return self.capacity if self.full else self.pos

    @property
    def size(self) -> int:
        return len(self)

    def reset(self) -> None:
        self.pos = 0
        self.full = False
        self.advantages = None
        self.returns = None

    # ------------------------------------------------------------------
    def _init_storage(self, sample: Dict[str, Any]) -> None:
        def alloc(t: torch.Tensor) -> torch.Tensor:
            return torch.zeros(
                (self.capacity, *t.shape), dtype=t.dtype, device=self.device
            )

        self._storage = tree_map(alloc, sample)

    @torch.no_grad()
    def add(
        self,
        obs,
        action,
        reward,
        done,
        value,
        log_prob,
        action_mask=None,
        hidden_state=None,
        **extras,
    ) -> None:
        if isinstance(obs, (tuple, list)):
            # tuple observation spaces are stored as index-keyed dicts; the
            # multi-input encoder accepts either form
            obs = {str(i): v for i, v in enumerate(obs)}
        data = {
            "obs": obs,
            "action": action,
            "reward": reward,
            "done": done,
            "value": value,
            "log_prob": log_prob,
        }
        if action_mask is not None:
            data["action_mask"] = action_mask
        if hidden_state is not None:
            data["hidden_state"] = hidden_state
        data.update({k: v for k, v in extras.items() if v is not None})
        data = tree_map(
            lambda t: t.detach() if isinstance(t, torch.Tensor) else t, to_tensor(data)
        )
        data = tree_map(lambda t: t.float() if t.dtype == torch.float64 else t, data)
        data = tree_map(lambda t: t.to(self.device), data)
        if self._storage is None:
            self._init_storage(data)
        if self.pos >= self.capacity:
            if not self.wrap_at_capacity:
                raise RuntimeError(
                    f"RolloutBuffer full ({self.capacity} steps); pass "
                    "wrap_at_capacity=True to overwrite as a ring"
                )
            self.pos = 0
        for key, val in data.items():
            _write(self._storage[key], val, self.pos)
        self.pos += 1
        if self.pos >= self.capacity:
            self.full = True

    # ------------------------------------------------------------------
    @torch.no_grad()
    def compute_returns_and_advantages(self, last_value, last_done=None) -> None:
        # ``last_done`` is accepted for backward compatibility but unused:
        # dones[t] = done-after-step-t already cuts the final step's bootstrap.
        T = len(self)
        sd = self._storage
        last_value = to_tensor(last_value).to(self.device).float().reshape(-1)
        values = sd["value"][:T].reshape(T, self.num_envs)
        rewards = sd["reward"][:T].reshape(T, self.num_envs)
        dones = sd["done"][:T].reshape(T, self.num_envs).float()
        lam = self.gae_lambda if self.use_gae else 1.0
        adv, ret = ops.gae_scan(
            rewards, values, dones, last_value, self.gamma, lam
        )
        if not self.use_gae:
            # plain discounted returns: lambda=1 GAE gives ret = discounted
            # rewards-to-go; advantage = ret - V
            adv = ret - values
        self.advantages = adv
        self.returns = ret

    # ------------------------------------------------------------------
    def get_tensor_batch(self) -> Dict[str, torch.Tensor]:
        """All data flattened to (T*N, ...)."""
        T = len(self)
        out = tree_map(lambda t: t[:T].reshape(T * self.num_envs, *t.shape[2:]), self._storage)
        if self.advantages is not None:
            out["advantages"] = self.advantages.reshape(-1)
            out["returns"] = self.returns.reshape(-1)
        return out

    def get_minibatches(
        self, batch_size: int, shuffle: bool = True
    ) -> Iterator[Dict[str, torch.Tensor]]:
        flat = self.get_tensor_batch()
        n = len(self) * self.num_envs
        idx = torch.randperm(n, device=self.device) if shuffle else torch.arange(n, device=self.device)
        for start in range(0, n, batch_size):
            mb_idx = idx[start : start + batch_size]
            yield tree_map(lambda t: t[mb_idx], flat)

    def get_sequence_minibatches(
        self, seq_len: Optional[int] = None, batch_size: int = 64,
        shuffle: bool = True, sequence_type: Optional[str] = None,
    ) -> Iterator[Dict[str, torch.Tensor]]:
        """(B, L, ...) contiguous same-env sequences for BPTT (recurrent PPO).

        ``sequence_type`` (reference typing.py:473 BPTTSequenceType) sets the
        start stride: "chunked" = non-overlapping (stride L), "maximum" = all
        overlapping windows (stride 1), "fifty_percent_overlap" = stride L/2.
        """
        seq_len = int(seq_len if seq_len is not None else (self.max_seq_len or 16))
        sequence_type = sequence_type or self.bptt_sequence_type
        stride = {"chunked": seq_len, "maximum": 1,
                  "fifty_percent_overlap": max(seq_len // 2, 1)}.get(sequence_type)
        if stride is None:
            raise ValueError(f"unknown BPTT sequence_type '{sequence_type}'")
        T = len(self)
        starts: List[Tuple[int, int]] = []
        for env in range(self.num_envs):
            for t0 in range(0, T - seq_len + 1, stride):
                starts.append((t0, env))
        order = np.random.permutation(len(starts)) if shuffle else np.arange(len(starts))
        flatstore = self._storage
        for i in range(0, len(starts), batch_size):
            chunk = [starts[j] for j in order[i : i + batch_size]]
            t0s = torch.tensor([c[0] for c in chunk], device=self.device)
            envs = torch.tensor([c[1] for c in chunk], device=self.device)
            t_idx = t0s.unsqueeze(1) + torch.arange(seq_len, device=self.device).unsqueeze(0)

            def gather(t: torch.Tensor) -> torch.Tensor:
                # t: (T, N, ...) -> (B, L, ...)
                return t[:T][t_idx, envs.unsqueeze(1).expand_as(t_idx)]

            out = tree_map(gather, flatstore)
            if self.advantages is not None:
                out["advantages"] = self.advantages[t_idx, envs.unsqueeze(1).expand_as(t_idx)]
                out["returns"] = self.returns[t_idx, envs.unsqueeze(1).expand_as(t_idx)]
            yield out


def _write(store, val, pos):
    if isinstance(store, dict):
        for k in store:
            _write(store[k], val[k], pos)
    else:
        store[pos] = val.to(store.device, store.dtype)
