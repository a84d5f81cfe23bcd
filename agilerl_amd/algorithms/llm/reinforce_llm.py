"""LLM REINFORCE with leave-one-out (RLOO) baseline.

Reference parity: ``agilerl/algorithms/reinforce_llm.py:66``.  Advantage
for completion i in its group: ``r_i - mean_{j != i} r_j`` =
``G/(G-1) * (r_i - group_mean)``; the policy gradient flows through the
token logprobs directly (no ratio/clipping), which is the CISPO kernel
path with unit weights.
"""

from __future__ import annotations

import torch

from ... import ops
from .grpo import GRPO

__all__ = ["ReinforceLLM"]


class ReinforceLLM(GRPO):
    CISPO = True  # fused kernel: loss = -w * adv * logp with w = clamp(ratio)

    def __init__(self, *args, gamma: float = 1.0, **kwargs):
        kwargs.setdefault("update_epochs", 1)
        kwargs.setdefault("clip_coef", 1e6)  # effectively unclipped weight ~ 1
        super().__init__(*args, **kwargs)
        self.algo = "ReinforceLLM"
        # reference reinforce_llm.py gamma: discounts turn rewards into the
        # return when the env emits per-turn rewards (1.0 = plain sum)
        self.gamma = float(gamma)

    def _calculate_advantages(self, rewards: torch.Tensor) -> torch.Tensor:
        G = self.group_size
        centered = ops.group_advantage(rewards.to(self.device).float(), G, scale=False)
        if G > 1:
            centered = centered * (G / (G - 1))  # leave-one-out rescale
        return centered
