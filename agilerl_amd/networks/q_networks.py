"""Q-value networks.

Reference parity: ``agilerl/networks/q_networks.py`` — QNetwork :22,
RainbowQNetwork :142 (noisy + dueling + C51 distributional),
ContinuousQNetwork :304.
"""

from __future__ import annotations

from typing import Any, Dict, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..modules.mlp import EvolvableMLP
from ..spaces import Box, Discrete, Space, flatdim
from .base import EvolvableNetwork

__all__ = ["QNetwork", "RainbowQNetwork", "ContinuousQNetwork"]


class QNetwork(EvolvableNetwork):
    """State -> Q(s, .) over discrete actions."""

    def __init__(
        self,
        observation_space: Space,
        action_space: Discrete,
        encoder_config: Optional[Dict[str, Any]] = None,
        head_config: Optional[Dict[str, Any]] = None,
        latent_dim: int = 64,
        device: str = "cpu",
        **net_kwargs,
    ):
        if not hasattr(action_space, "n"):
            raise TypeError(
                f"QNetwork requires a discrete action space, got "
                f"{type(action_space).__name__} (use DDPG/TD3/PPO for Box actions)"
            )
        self.action_space = action_space
        super().__init__(
            observation_space,
            num_outputs=action_space.n,
            encoder_config=encoder_config,
            head_config=head_config,
            latent_dim=latent_dim,
            device=device,
            **net_kwargs,
        )


class RainbowQNetwork(EvolvableNetwork):
    """Dueling distributional (C51) noisy Q network.

    Value/advantage streams over ``num_atoms`` support atoms; forward
    returns per-action Q values, :meth:`dist` returns the atom
    distribution (B, A, num_atoms).  The dueling+softmax combine is a HIP
    fusion target.
    """

    def __init__(
        self,
        observation_space: Space,
        action_space: Discrete,
        encoder_config: Optional[Dict[str, Any]] = None,
        head_config: Optional[Dict[str, Any]] = None,
        latent_dim: int = 64,
        num_atoms: int = 51,
        v_min: float = -10.0,
        v_max: float = 10.0,
        noise_std: float = 0.5,
        support=None,
        device: str = "cpu",
        **net_kwargs,
    ):
        self.action_space = action_space
        self.num_atoms = int(num_atoms)
        self.v_min = float(v_min)
        self.v_max = float(v_max)
        self.noise_std = noise_std
        head_config = dict(head_config or {"hidden_size": [64]})
        head_config.setdefault("noisy", True)
        head_config.setdefault("noise_std", noise_std)
        head_config.setdefault("init_layers", False)
        super().__init__(
            observation_space,
            num_outputs=action_space.n * num_atoms,
            encoder_config=encoder_config,
            head_config=head_config,
            latent_dim=latent_dim,
            device=device,
            **net_kwargs,
        )
        self.register_buffer(
            "support",
            torch.as_tensor(support, dtype=torch.float32, device=device)
            if support is not None
            else torch.linspace(self.v_min, self.v_max, self.num_atoms, device=device)
        )

    def _build_head(self):
        # advantage stream = self.head_net (via parent); value stream built here
        cfg = dict(self.head_config or {"hidden_size": [64]})
        cfg.pop("arch", None)
        self.value_net = EvolvableMLP(
            num_inputs=self.latent_dim, num_outputs=self.num_atoms, device=self.device, **cfg
        )
        return EvolvableMLP(
            num_inputs=self.latent_dim,
            num_outputs=self.action_space.n * self.num_atoms,
            device=self.device,
            **cfg,
        )

    def dist(self, obs) -> torch.Tensor:
        """(B, A, num_atoms) atom probabilities."""
        feats = self.extract_features(obs)
        adv = self.head_net(feats).view(-1, self.action_space.n, self.num_atoms)
        val = self.value_net(feats).view(-1, 1, self.num_atoms)
        q_atoms = val + adv - adv.mean(dim=1, keepdim=True)
        return F.softmax(q_atoms, dim=-1).clamp(min=1e-8)

    def forward(self, obs) -> torch.Tensor:
        return (self.dist(obs) * self.support.view(1, 1, -1)).sum(-1)

    def reset_noise(self) -> None:
        super().reset_noise()
        self.value_net.reset_noise()

    @property
    def mutation_methods(self):
        # value stream mirrors the advantage stream; mutations replay on both
        return super().mutation_methods

    def apply_mutation(self, name: str, **choices):
        result = super().apply_mutation(name, **choices)
        if name.startswith("head."):
            merged = {**choices, **(result or {})}
            self.value_net.apply_mutation(name[len("head.") :], **merged)
        elif name in ("add_latent_node", "remove_latent_node"):
            self.value_net.num_inputs = self.latent_dim
            self.value_net.recreate_network()
        return result


class ContinuousQNetwork(EvolvableNetwork):
    """(state, action) -> scalar Q. Critic for DDPG/TD3/MADDPG/MATD3."""

    def __init__(
        self,
        observation_space: Space,
        action_space: Box,
        encoder_config: Optional[Dict[str, Any]] = None,
        head_config: Optional[Dict[str, Any]] = None,
        latent_dim: int = 64,
        action_dim: Optional[int] = None,
        device: str = "cpu",
        **net_kwargs,
    ):
        self.action_space = action_space
        self.action_dim = int(action_dim) if action_dim is not None else flatdim(action_space)
        super().__init__(
            observation_space,
            num_outputs=1,
            encoder_config=encoder_config,
            head_config=head_config,
            latent_dim=latent_dim,
            device=device,
            **net_kwargs,
        )

    def _build_head(self):
        cfg = dict(self.head_config or {"hidden_size": [64]})
        cfg.pop("arch", None)
        return EvolvableMLP(
            num_inputs=self.latent_dim + self.action_dim,
            num_outputs=1,
            device=self.device,
            **cfg,
        )

    def forward(self, obs, action: torch.Tensor) -> torch.Tensor:
        feats = self.extract_features(obs)
        action = action.to(feats.device).float()
        if action.dim() == 1:
            action = action.unsqueeze(0)
        return self.head_net(torch.cat([feats, action.reshape(feats.shape[0], -1)], dim=-1))

    def _resize_latent(self, new_dim: int, resize_encoder: bool = True) -> None:
        import numpy as np

        new_dim = int(np.clip(new_dim, self.MIN_LATENT, self.MAX_LATENT))
        if new_dim == self.latent_dim:
            return
        self.latent_dim = new_dim
        if resize_encoder:
            self.encoder.num_outputs = new_dim
            self.encoder.recreate_network()
        self.head_net.num_inputs = new_dim + self.action_dim
        self.head_net.recreate_network()
