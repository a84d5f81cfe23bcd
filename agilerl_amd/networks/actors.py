"""Actor networks.

Reference parity: ``agilerl/networks/actors.py`` — DeterministicActor :43
(with action rescaling for Box spaces), StochasticActor :262 (wraps the
distribution layer).
"""

from __future__ import annotations

from typing import Any, Dict, Optional, Tuple

import torch

from ..spaces import Box, Discrete, Space, flatdim
from .base import EvolvableNetwork
from .distributions import ActionDistribution

__all__ = ["DeterministicActor", "StochasticActor"]


class DeterministicActor(EvolvableNetwork):
    """State -> action (tanh-bounded, rescaled to the Box bounds)."""

    def __init__(
        self,
        observation_space: Space,
        action_space: Space,
        encoder_config: Optional[Dict[str, Any]] = None,
        head_config: Optional[Dict[str, Any]] = None,
        latent_dim: int = 64,
        device: str = "cpu",
        **net_kwargs,
    ):
        self.action_space = action_space
        head_config = dict(head_config or {"hidden_size": [64]})
        if isinstance(action_space, Box):
            head_config.setdefault("output_activation", "Tanh")
            num_outputs = flatdim(action_space)
        elif isinstance(action_space, Discrete):
            head_config.setdefault("output_activation", "GumbelSoftmax")
            num_outputs = action_space.n
        else:
            num_outputs = flatdim(action_space)
        super().__init__(
            observation_space,
            num_outputs=num_outputs,
            encoder_config=encoder_config,
            head_config=head_config,
            latent_dim=latent_dim,
            device=device,
            **net_kwargs,
        )
        if isinstance(action_space, Box):
            low = torch.as_tensor(action_space.low, dtype=torch.float32)
            high = torch.as_tensor(action_space.high, dtype=torch.float32)
            finite = torch.isfinite(low) & torch.isfinite(high)
            low = torch.where(finite, low, torch.full_like(low, -1.0))
            high = torch.where(finite, high, torch.full_like(high, 1.0))
            self.register_buffer("action_low", low.to(device))
            self.register_buffer("action_high", high.to(device))

    def forward(self, obs) -> torch.Tensor:
        out = super().forward(obs)
        if isinstance(self.action_space, Box):
            out = self.action_low + (out + 1.0) * 0.5 * (self.action_high - self.action_low)
        return out

    def raw_forward(self, obs) -> torch.Tensor:
        """Unrescaled (tanh / gumbel) output."""
        return EvolvableNetwork.forward(self, obs)


class StochasticActor(EvolvableNetwork):
    """State -> action distribution (PPO/IPPO and co).

    ``forward`` returns the raw head output (logits / mean); use
    :meth:`action_dist`, :meth:`sample`, :meth:`evaluate_actions`.
    """

    def __init__(
        self,
        observation_space: Space,
        action_space: Space,
        encoder_config: Optional[Dict[str, Any]] = None,
        head_config: Optional[Dict[str, Any]] = None,
        latent_dim: int = 64,
        log_std_init: float = 0.0,
        action_std_init: Optional[float] = None,
        squash_output: bool = False,
        device: str = "cpu",
        **net_kwargs,
    ):
        if action_std_init is not None:
            # reference actors.py: initial LOG std despite the name
            log_std_init = float(action_std_init)
        self.action_space = action_space
        dist = ActionDistribution(action_space, log_std_init=log_std_init, squash=squash_output)
        super().__init__(
            observation_space,
            num_outputs=dist.head_output_size,
            encoder_config=encoder_config,
            head_config=head_config,
            latent_dim=latent_dim,
            device=device,
            **net_kwargs,
        )
        self.dist_layer = dist.to(device)

    def sample(
        self, obs, action_mask: Optional[torch.Tensor] = None
    ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
        head_out = super().forward(obs)
        return self.dist_layer.sample(head_out, action_mask)

    def evaluate_actions(
        self, obs, actions: torch.Tensor, action_mask: Optional[torch.Tensor] = None
    ) -> Tuple[torch.Tensor, torch.Tensor]:
        head_out = super().forward(obs)
        return self.dist_layer.log_prob_entropy(head_out, actions, action_mask)

    def deterministic_action(self, obs, action_mask: Optional[torch.Tensor] = None) -> torch.Tensor:
        head_out = super().forward(obs)
        return self.dist_layer.mode(head_out, action_mask)
