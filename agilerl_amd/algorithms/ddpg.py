"""DDPG (deterministic actor-critic, OU/Gaussian exploration noise).

Reference parity: ``agilerl/algorithms/ddpg.py:50`` — per-env noise state
with ``reset_action_noise`` (used by the off-policy loop at
``train_off_policy.py:387``), polyak targets (HIP-fused).
"""

from __future__ import annotations

from typing import Any, Dict, Optional

import numpy as np
import torch
import torch.nn.functional as F

from .. import ops
from ..networks.actors import DeterministicActor
from ..networks.q_networks import ContinuousQNetwork
from ..spaces import Box, Space
from .core.base import RLAlgorithm
from .core.optimizer_wrapper import OptimizerWrapper
from .core.registry import HyperparameterConfig, NetworkGroup, OptimizerConfig, RLParameter

__all__ = ["DDPG"]


def default_hp_config() -> HyperparameterConfig:
    return HyperparameterConfig(
        lr_actor=RLParameter(min=1e-5, max=1e-2),
        lr_critic=RLParameter(min=1e-5, max=1e-2),
        batch_size=RLParameter(min=16, max=1024, dtype=int),
        learn_step=RLParameter(min=1, max=16, dtype=int),
    )


class DDPG(RLAlgorithm):
    def __init__(
        self,
        observation_space: Space,
        action_space: Box,
        index: int = 0,
        hp_config: Optional[HyperparameterConfig] = None,
        net_config: Optional[Dict[str, Any]] = None,
        head_config: Optional[Dict[str, Any]] = None,
        batch_size: int = 64,
        lr_actor: float = 1e-4,
        lr_critic: float = 1e-3,
        learn_step: int = 5,
        gamma: float = 0.99,
        tau: float = 1e-3,
        policy_freq: int = 1,
        O_U_noise: bool = True,
        expl_noise: float = 0.1,
        mean_noise: float = 0.0,
        theta: float = 0.15,
        dt: float = 1e-2,
        vect_noise_dim: int = 1,
        latent_dim: int = 64,
        share_encoders: bool = False,
        actor_network=None,
        critic_network=None,
        device: str = "cpu",
        **kwargs,
    ):
        super().__init__(
            observation_space, action_space, index=index, learn_step=learn_step,
            device=device, hp_config=hp_config or default_hp_config(), name="DDPG",
        )
        self._accept_compat_kwargs(**kwargs)
        self.batch_size = int(batch_size)
        self.lr_actor = float(lr_actor)
        self.lr_critic = float(lr_critic)
        self.gamma = float(gamma)
        self.tau = float(tau)
        self.policy_freq = int(policy_freq)
        self.O_U_noise = O_U_noise
        self.expl_noise = expl_noise
        self.mean_noise = mean_noise
        self.theta = theta
        self.dt = dt
        self.net_config = net_config
        self.latent_dim = latent_dim
        self._learn_counter = 0
        # reference ddpg.py:119 vect_noise_dim: per-env OU noise rows,
        # presized here and auto-resized on the first batched get_action
        self.vect_noise_dim = int(vect_noise_dim)
        self._ou_state: Optional[np.ndarray] = (
            np.zeros((self.vect_noise_dim, self.action_dim)) if vect_noise_dim > 1 else None
        )

        if actor_network is not None:
            # user-supplied policy net (reference ddpg.py actor_network);
            # must map preprocessed obs -> in-range actions
            from ..networks.base import CustomNetworkAdapter

            self.actor = CustomNetworkAdapter(actor_network, observation_space,
                                              action_space=action_space, device=device)
        else:
            self.actor = DeterministicActor(
                observation_space, action_space, encoder_config=net_config,
                head_config=head_config, latent_dim=latent_dim, device=device,
            )
        self.actor_target = self.actor.clone()
        if critic_network is not None:
            # user-supplied (state, action) -> Q net (reference ddpg.py:136)
            from ..networks.base import CustomQAdapter

            self.critic = CustomQAdapter(critic_network, observation_space,
                                         action_space=action_space, device=device)
        else:
            self.critic = ContinuousQNetwork(
                observation_space, action_space, encoder_config=net_config,
                head_config=head_config, latent_dim=latent_dim, device=device,
            )
        self.critic_target = self.critic.clone()
        for net in (self.actor_target, self.critic_target):
            for p in net.parameters():
                p.requires_grad = False

        # reference ddpg.py:315: critic's encoder is pinned to a detached copy
        # of the actor's (trained only through the actor loss); re-pinned after
        # every mutation via the hook below
        self.share_encoders = bool(share_encoders)
        if self.share_encoders:
            if not (hasattr(self.actor, "encoder") and hasattr(self.critic, "encoder")):
                import warnings

                warnings.warn("share_encoders disabled: actor/critic has no encoder")
                self.share_encoders = False
            else:
                self.share_encoder_parameters()

        self.actor_optimizer = OptimizerWrapper(torch.optim.Adam, [self.actor], lr=self.lr_actor)
        self.critic_optimizer = OptimizerWrapper(torch.optim.Adam, [self.critic], lr=self.lr_critic)

        self.register_network_group(
            NetworkGroup(eval_network="actor", shared_networks=["actor_target"], policy=True)
        )
        self.register_network_group(
            NetworkGroup(eval_network="critic", shared_networks=["critic_target"])
        )
        self.register_optimizer(
            OptimizerConfig(name="actor_optimizer", networks=["actor"], lr_name="lr_actor")
        )
        self.register_optimizer(
            OptimizerConfig(name="critic_optimizer", networks=["critic"], lr_name="lr_critic")
        )
        self.register_mutation_hook("_sync_targets_after_mutation")
        if self.share_encoders:
            self.register_mutation_hook("share_encoder_parameters")

    def _sync_targets_after_mutation(self) -> None:
        self.actor_target.load_state_dict(self.actor.state_dict())
        self.critic_target.load_state_dict(self.critic.state_dict())
        for net in (self.actor_target, self.critic_target):
            for p in net.parameters():
                p.requires_grad = False

    def share_encoder_parameters(self) -> None:
        """Copy the actor's encoder weights into the critic (+ target) and
        freeze them there, so only the actor loss trains the encoder."""
        state = {k: v.detach().clone() for k, v in self.actor.encoder.state_dict().items()}
        for net in (self.critic, self.critic_target):
            enc = getattr(net, "encoder", None)
            if enc is not None:
                enc.load_state_dict(state)
        for p in self.critic.encoder.parameters():
            p.requires_grad = False

    # ------------------------------------------------------------------
    @property
    def action_dim(self) -> int:
        return int(np.prod(self.action_space.shape))

    def reset_action_noise(self, indices=None) -> None:
        """Reset OU noise state (per finished env)."""
        if self._ou_state is None:
            return
        if indices is None:
            self._ou_state[:] = 0.0
        else:
            self._ou_state[np.asarray(indices)] = 0.0

    def _sample_noise(self, n: int) -> np.ndarray:
        if self._ou_state is None or self._ou_state.shape[0] != n:
            self._ou_state = np.zeros((n, self.action_dim))
        if self.O_U_noise:
            dx = self.theta * (self.mean_noise - self._ou_state) * self.dt + self.expl_noise * np.sqrt(
                self.dt
            ) * np.random.randn(n, self.action_dim)
            self._ou_state = self._ou_state + dx
            return self._ou_state
        return np.random.normal(self.mean_noise, self.expl_noise, size=(n, self.action_dim))

    def get_action(self, obs, training: bool = True, **kwargs) -> np.ndarray:
        with torch.no_grad():
            action = self.actor(self.actor.preprocess(obs)).cpu().numpy()
        if training:
            action = action + self._sample_noise(action.shape[0])
        low, high = self.action_space.low, self.action_space.high
        return np.clip(action, low, high)

    # ------------------------------------------------------------------
    def learn(self, experiences: Dict[str, torch.Tensor]) -> float:
        obs = experiences["obs"]
        actions = experiences["action"].to(self.device).float()
        rewards = experiences["reward"].to(self.device).float().reshape(-1, 1)
        next_obs = experiences["next_obs"]
        dones = experiences["done"].to(self.device).float().reshape(-1, 1)

        with torch.no_grad():
            next_actions = self.actor_target(self.actor_target.preprocess(next_obs))
            q_next = self.critic_target(self.critic_target.preprocess(next_obs), next_actions)
            target = rewards + (1.0 - dones) * self.gamma * q_next

        q = self.critic(self.critic.preprocess(obs), actions)
        critic_loss = F.mse_loss(q, target)
        self.critic_optimizer.zero_grad()
        critic_loss.backward()
        self.critic_optimizer.step()

        self._learn_counter += 1
        actor_loss_val = 0.0
        if self._learn_counter % self.policy_freq == 0:
            pre = self.actor.preprocess(obs)
            actor_loss = -self.critic(self.critic.preprocess(obs), self.actor(pre)).mean()
            self.actor_optimizer.zero_grad()
            actor_loss.backward()
            self.actor_optimizer.step()
            actor_loss_val = float(actor_loss.detach())
            self.soft_update()
        return float(critic_loss.detach()) + actor_loss_val * 0.0

    def soft_update(self) -> None:
        ops.polyak_update_(
            list(self.actor_target.parameters()), list(self.actor.parameters()), self.tau
        )
        ops.polyak_update_(
            list(self.critic_target.parameters()), list(self.critic.parameters()), self.tau
        )
