"""Action distributions with masking support.

Reference parity: ``agilerl/networks/distributions.py`` (TorchDistribution
:40 — Categorical/Normal/Bernoulli + action masking; EvolvableDistribution
:119).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.nn as nn
from torch.distributions import Bernoulli, Categorical, Normal

from ..spaces import Box, Discrete, MultiBinary, MultiDiscrete, Space

__all__ = ["ActionDistribution", "EvolvableDistribution"]


def _masked_logits(logits: torch.Tensor, mask: Optional[torch.Tensor]) -> torch.Tensor:
    if mask is None:
        return logits
    mask = mask.to(dtype=torch.bool, device=logits.device)
    return logits.masked_fill(~mask, torch.finfo(logits.dtype).min)


class ActionDistribution(nn.Module):
    """Maps network head outputs -> a torch distribution for the action space.

    - Discrete        -> Categorical over n logits
    - MultiDiscrete   -> independent Categoricals (split logits)
    - MultiBinary     -> Bernoulli over n logits
    - Box             -> diagonal Normal; state-independent learnable log_std;
                         optional tanh squashing (SAC-style) via `squash`.
    """

    def __init__(
        self,
        action_space: Space,
        log_std_init: float = 0.0,
        squash: bool = False,
    ):
        super().__init__()
        self.action_space = action_space
        self.squash = squash
        if isinstance(action_space, Box):
            self.action_dim = int(torch.tensor(action_space.shape).prod().item())
            self.log_std = nn.Parameter(torch.ones(self.action_dim) * log_std_init)
        elif isinstance(action_space, Discrete):
            self.action_dim = action_space.n
        elif isinstance(action_space, MultiDiscrete):
            self.action_dim = int(sum(action_space.nvec))
        elif isinstance(action_space, MultiBinary):
            self.action_dim = int(torch.tensor(action_space.shape).prod().item())
        else:
            raise TypeError(f"Unsupported action space {type(action_space)}")

    @property
    def head_output_size(self) -> int:
        return self.action_dim

    def distribution(self, head_out: torch.Tensor, action_mask: Optional[torch.Tensor] = None):
        space = self.action_space
        if isinstance(space, Discrete):
            return Categorical(logits=_masked_logits(head_out, action_mask))
        if isinstance(space, MultiDiscrete):
            splits = torch.split(head_out, list(space.nvec), dim=-1)
            masks = (
                torch.split(action_mask, list(space.nvec), dim=-1)
                if action_mask is not None
                else [None] * len(splits)
            )
            return [Categorical(logits=_masked_logits(s, m)) for s, m in zip(splits, masks)]
        if isinstance(space, MultiBinary):
            return Bernoulli(logits=head_out)
        # Box
        log_std = self.log_std.clamp(-20.0, 2.0)
        return Normal(head_out, log_std.exp().expand_as(head_out))

    def sample(
        self, head_out: torch.Tensor, action_mask: Optional[torch.Tensor] = None
    ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
        """Returns (action, log_prob, entropy)."""
        dist = self.distribution(head_out, action_mask)
        if isinstance(dist, list):  # MultiDiscrete
            actions = torch.stack([d.sample() for d in dist], dim=-1)
            logp = torch.stack([d.log_prob(actions[..., i]) for i, d in enumerate(dist)], dim=-1).sum(-1)
            ent = torch.stack([d.entropy() for d in dist], dim=-1).sum(-1)
            return actions, logp, ent
        if isinstance(dist, Normal):
            raw = dist.rsample()
            if self.squash:
                action = torch.tanh(raw)
                logp = dist.log_prob(raw) - torch.log(1 - action.pow(2) + 1e-6)
                return action, logp.sum(-1), dist.entropy().sum(-1)
            return raw, dist.log_prob(raw).sum(-1), dist.entropy().sum(-1)
        if isinstance(dist, Bernoulli):
            action = dist.sample()
            return action, dist.log_prob(action).sum(-1), dist.entropy().sum(-1)
        action = dist.sample()
        return action, dist.log_prob(action), dist.entropy()

    def log_prob_entropy(
        self,
        head_out: torch.Tensor,
        action: torch.Tensor,
        action_mask: Optional[torch.Tensor] = None,
    ) -> Tuple[torch.Tensor, torch.Tensor]:
        dist = self.distribution(head_out, action_mask)
        if isinstance(dist, list):
            logp = torch.stack(
                [d.log_prob(action[..., i].long()) for i, d in enumerate(dist)], dim=-1
            ).sum(-1)
            ent = torch.stack([d.entropy() for d in dist], dim=-1).sum(-1)
            return logp, ent
        if isinstance(dist, Normal):
            if self.squash:
                raw = torch.atanh(action.clamp(-1 + 1e-6, 1 - 1e-6))
                logp = dist.log_prob(raw) - torch.log(1 - action.pow(2) + 1e-6)
                return logp.sum(-1), dist.entropy().sum(-1)
            return dist.log_prob(action).sum(-1), dist.entropy().sum(-1)
        if isinstance(dist, Bernoulli):
            return dist.log_prob(action).sum(-1), dist.entropy().sum(-1)
        return dist.log_prob(action.long().squeeze(-1) if action.dim() > 1 else action.long()), dist.entropy()

    def mode(self, head_out: torch.Tensor, action_mask: Optional[torch.Tensor] = None) -> torch.Tensor:
        dist = self.distribution(head_out, action_mask)
        if isinstance(dist, list):
            return torch.stack([d.logits.argmax(-1) for d in dist], dim=-1)
        if isinstance(dist, Normal):
            return torch.tanh(dist.mean) if self.squash else dist.mean
        if isinstance(dist, Bernoulli):
            return (dist.logits > 0).float()
        return dist.logits.argmax(-1)


# reference class name (agilerl/networks/distributions.py:119)
EvolvableDistribution = ActionDistribution
