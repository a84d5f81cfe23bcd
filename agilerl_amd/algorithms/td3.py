"""TD3 (twin critics, delayed policy updates, target policy smoothing).

Reference parity: ``agilerl/algorithms/td3.py:50``.
"""

from __future__ import annotations

from typing import Any, Dict, Optional

import torch
import torch.nn.functional as F

from .. import ops
from ..networks.q_networks import ContinuousQNetwork
from ..spaces import Box, Space
from .core.registry import HyperparameterConfig, NetworkGroup, OptimizerConfig
from .core.optimizer_wrapper import OptimizerWrapper
from .ddpg import DDPG

__all__ = ["TD3"]


class TD3(DDPG):
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
        tau: float = 5e-3,
        policy_freq: int = 2,
        policy_noise: float = 0.2,
        noise_clip: float = 0.5,
        O_U_noise: bool = True,
        expl_noise: float = 0.1,
        mean_noise: float = 0.0,
        theta: float = 0.15,
        dt: float = 1e-2,
        latent_dim: int = 64,
        share_encoders: bool = False,
        actor_network=None,
        critic_networks=None,
        device: str = "cpu",
        **kwargs,
    ):
        if critic_networks is not None and len(critic_networks) != 2:
            raise ValueError("TD3 critic_networks must be a list of two nets")
        super().__init__(
            observation_space, action_space, index=index, hp_config=hp_config,
            net_config=net_config, head_config=head_config, batch_size=batch_size,
            lr_actor=lr_actor, lr_critic=lr_critic, learn_step=learn_step,
            gamma=gamma, tau=tau, policy_freq=policy_freq, O_U_noise=O_U_noise,
            expl_noise=expl_noise, mean_noise=mean_noise, theta=theta, dt=dt,
            latent_dim=latent_dim, share_encoders=share_encoders,
            actor_network=actor_network,
            critic_network=None if critic_networks is None else critic_networks[0],
            device=device,
            **kwargs,
        )
        self.algo = "TD3"
        self.policy_noise = policy_noise
        self.noise_clip = noise_clip

        # second critic (twin)
        if critic_networks is not None:
            from ..networks.base import CustomQAdapter

            self.critic_2 = CustomQAdapter(critic_networks[1], observation_space,
                                           action_space=action_space, device=device)
        else:
            self.critic_2 = ContinuousQNetwork(
                observation_space, action_space, encoder_config=net_config,
                head_config=head_config, latent_dim=latent_dim, device=device,
            )
        self.critic_2_target = self.critic_2.clone()
        for p in self.critic_2_target.parameters():
            p.requires_grad = False
        self.critic_2_optimizer = OptimizerWrapper(
            torch.optim.Adam, [self.critic_2], lr=self.lr_critic
        )
        self.register_network_group(
            NetworkGroup(eval_network="critic_2", shared_networks=["critic_2_target"])
        )
        self.register_optimizer(
            OptimizerConfig(name="critic_2_optimizer", networks=["critic_2"], lr_name="lr_critic")
        )
        self.register_mutation_hook("_sync_twin_after_mutation")
        if self.share_encoders:
            self.share_encoder_parameters()  # now also pins critic_2

    def _sync_twin_after_mutation(self) -> None:
        self.critic_2_target.load_state_dict(self.critic_2.state_dict())
        for p in self.critic_2_target.parameters():
            p.requires_grad = False

    def share_encoder_parameters(self) -> None:
        super().share_encoder_parameters()
        if not hasattr(self, "critic_2"):  # called from DDPG.__init__ pre-twin
            return
        state = {k: v.detach().clone() for k, v in self.actor.encoder.state_dict().items()}
        for net in (self.critic_2, self.critic_2_target):
            enc = getattr(net, "encoder", None)
            if enc is not None:
                enc.load_state_dict(state)
        for p in self.critic_2.encoder.parameters():
            p.requires_grad = False

    # ------------------------------------------------------------------
    def learn(self, experiences: Dict[str, torch.Tensor]) -> float:
        obs = experiences["obs"]
        actions = experiences["action"].to(self.device).float()
        rewards = experiences["reward"].to(self.device).float().reshape(-1, 1)
        next_obs = experiences["next_obs"]
        dones = experiences["done"].to(self.device).float().reshape(-1, 1)

        with torch.no_grad():
            next_actions = self.actor_target(self.actor_target.preprocess(next_obs))
            noise = (torch.randn_like(next_actions) * self.policy_noise).clamp(
                -self.noise_clip, self.noise_clip
            )
            low = self.actor.action_low
            high = self.actor.action_high
            next_actions = (next_actions + noise).clamp(low, high)
            pre_next = self.critic_target.preprocess(next_obs)
            q1 = self.critic_target(pre_next, next_actions)
            q2 = self.critic_2_target(self.critic_2_target.preprocess(next_obs), next_actions)
            target = rewards + (1.0 - dones) * self.gamma * torch.minimum(q1, q2)

        pre = self.critic.preprocess(obs)
        critic_loss = F.mse_loss(self.critic(pre, actions), target)
        self.critic_optimizer.zero_grad()
        critic_loss.backward()
        self.critic_optimizer.step()

        critic2_loss = F.mse_loss(
            self.critic_2(self.critic_2.preprocess(obs), actions), target
        )
        self.critic_2_optimizer.zero_grad()
        critic2_loss.backward()
        self.critic_2_optimizer.step()

        self._learn_counter += 1
        if self._learn_counter % self.policy_freq == 0:
            actor_loss = -self.critic(
                self.critic.preprocess(obs), self.actor(self.actor.preprocess(obs))
            ).mean()
            self.actor_optimizer.zero_grad()
            actor_loss.backward()
            self.actor_optimizer.step()
            self.soft_update()
        return float(critic_loss.detach())

    def soft_update(self) -> None:
        super().soft_update()
        ops.polyak_update_(
            list(self.critic_2_target.parameters()), list(self.critic_2.parameters()), self.tau
        )
