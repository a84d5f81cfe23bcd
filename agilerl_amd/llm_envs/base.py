"""LLM environment base + experience assembly.

Reference parity: ``agilerl/llm_envs/base.py:93`` (HuggingFaceGym — HF
dataset + rank-sharded DataLoader, chat templating) — reshaped for this
framework: an LLM env yields *token batches* ready for the model
(prompts repeated ``group_size`` times), scores generated sequences, and
the loop assembles the (ids, action_mask, rewards) experiences dict the
GRPO-family ``learn()`` consumes.
"""

from __future__ import annotations

from typing import Dict

import numpy as np
import torch

__all__ = ["LLMEnvBase", "make_grpo_experiences"]


class LLMEnvBase:
    """Interface: reset() -> prompt batch; score(sequences) -> rewards."""

    group_size: int = 1
    prompt_len: int = 0  # padded prompt length of the last reset batch

    def reset(self) -> Dict[str, torch.Tensor]:
        raise NotImplementedError

    def score(self, sequences: torch.Tensor) -> np.ndarray:
        raise NotImplementedError

    def step(self, sequences: torch.Tensor):
        """Single-turn default: scoring ends the episode."""
        rewards = self.score(sequences)
        return rewards, True


def make_grpo_experiences(
    env: LLMEnvBase,
    sequences: torch.Tensor,
    rewards: np.ndarray,
    pad_token_id: int = 0,
) -> Dict[str, torch.Tensor]:
    """Assemble the GRPO learn() payload from generated sequences.

    ``action_mask[b, j]`` marks target position j (predicting token j+1)
    as a completion token: j+1 >= prompt_len and ids[b, j+1] != pad.
    """
    ids = sequences
    B, T = ids.shape
    P = env.prompt_len
    attention_mask = (ids != pad_token_id).long()
    # left-padded prompts: everything from position P on is completion
    pos = torch.arange(T - 1, device=ids.device).unsqueeze(0)
    action_mask = (pos + 1 >= P) & (ids[:, 1:] != pad_token_id)
    return {
        "ids": ids,
        "attention_mask": attention_mask,
        "action_mask": action_mask.float(),
        "rewards": torch.as_tensor(np.asarray(rewards), dtype=torch.float32),
    }
