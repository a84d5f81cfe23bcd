"""MATD3 — MADDPG with twin delayed centralized critics.

Reference parity: ``agilerl/algorithms/matd3.py:62``.
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional

import numpy as np
import torch
import torch.nn.functional as F

from .. import ops
from ..modules.base import ModuleDict
from ..networks.q_networks import ContinuousQNetwork
from ..spaces import Box, Space
from .core.optimizer_wrapper import OptimizerWrapper
from .core.registry import HyperparameterConfig, NetworkGroup, OptimizerConfig
from .maddpg import MADDPG

__all__ = ["MATD3"]


class MATD3(MADDPG):
    def __init__(
        self,
        observation_spaces: Dict[str, Space],
        action_spaces: Dict[str, Space],
        agent_ids: Optional[List[str]] = None,
        index: int = 0,
        hp_config: Optional[HyperparameterConfig] = None,
        net_config: Optional[Dict[str, Any]] = None,
        head_config: Optional[Dict[str, Any]] = None,
        batch_size: int = 64,
        lr_actor: float = 1e-4,
        lr_critic: float = 1e-3,
        learn_step: int = 5,
        gamma: float = 0.95,
        tau: float = 1e-2,
        policy_freq: int = 2,
        O_U_noise: bool = True,
        expl_noise: float = 0.1,
        mean_noise: float = 0.0,
        theta: float = 0.15,
        dt: float = 1e-2,
        latent_dim: int = 64,
        actor_networks: Optional[Dict[str, Any]] = None,
        critic_networks: Optional[Dict[str, Any]] = None,
        device: str = "cpu",
        **kwargs,
    ):
        # critic_networks: {agent_id: (critic_1, critic_2)} custom twin
        # centralized critics (reference matd3.py critic_networks)
        first = second = None
        if critic_networks is not None:
            bad = [a for a, v in critic_networks.items()
                   if not (isinstance(v, (list, tuple)) and len(v) == 2)]
            if bad:
                raise ValueError(
                    f"MATD3 critic_networks values must be (critic_1, critic_2) pairs; got bad entries for {bad}"
                )
            first = {a: v[0] for a, v in critic_networks.items()}
            second = {a: v[1] for a, v in critic_networks.items()}
        super().__init__(
            observation_spaces, action_spaces, agent_ids=agent_ids, index=index,
            hp_config=hp_config, net_config=net_config, head_config=head_config,
            batch_size=batch_size, lr_actor=lr_actor, lr_critic=lr_critic,
            learn_step=learn_step, gamma=gamma, tau=tau, O_U_noise=O_U_noise,
            expl_noise=expl_noise, mean_noise=mean_noise, theta=theta, dt=dt,
            latent_dim=latent_dim, actor_networks=actor_networks,
            critic_networks=first, device=device,
            **kwargs,
        )
        self.algo = "MATD3"
        self.policy_freq = int(policy_freq)
        self._learn_counter = 0

        joint_space = Box(-np.inf, np.inf, (self.joint_obs_dim,))

        def _make_critic_2(aid):
            if second is not None and aid in second:
                from ..networks.base import CustomQAdapter

                return CustomQAdapter(
                    second[aid], joint_space,
                    action_space=Box(-1.0, 1.0, (self.joint_action_dim,)),
                    device=device,
                )
            return ContinuousQNetwork(
                joint_space, Box(-1.0, 1.0, (self.joint_action_dim,)),
                encoder_config=net_config, head_config=head_config,
                latent_dim=latent_dim, action_dim=self.joint_action_dim, device=device,
            )

        self.critics_2 = ModuleDict(
            {aid: _make_critic_2(aid) for aid in self.agent_ids},
            device=device,
        )
        self.critic_2_targets = self.critics_2.clone()
        for p in self.critic_2_targets.parameters():
            p.requires_grad = False
        self.critic_2_optimizer = OptimizerWrapper(
            torch.optim.Adam, [self.critics_2], lr=self.lr_critic, multiagent=True
        )
        self.register_network_group(
            NetworkGroup(eval_network="critics_2", shared_networks=["critic_2_targets"], multiagent=True)
        )
        self.register_optimizer(
            OptimizerConfig(name="critic_2_optimizer", networks=["critics_2"], lr_name="lr_critic")
        )
        self.register_mutation_hook("_sync_twins_after_mutation")

    def _sync_twins_after_mutation(self) -> None:
        self.critic_2_targets.load_state_dict(self.critics_2.state_dict())
        for p in self.critic_2_targets.parameters():
            p.requires_grad = False

    # ------------------------------------------------------------------
    def learn(self, experiences: Dict[str, Dict[str, torch.Tensor]]) -> float:
        obs = self._to_dev(experiences["obs"])
        actions = self._to_dev(experiences["action"])
        rewards = self._to_dev(experiences["reward"])
        next_obs = self._to_dev(experiences["next_obs"])
        dones = self._to_dev(experiences["done"])

        joint_obs = self._joint(obs)
        joint_actions = self._joint(actions)
        with torch.no_grad():
            next_raw = {}
            for aid in self.agent_ids:
                tgt = self.actor_targets[aid]
                tgt.train()
                next_raw[aid] = tgt(tgt.preprocess(next_obs[aid]))
            joint_next_obs = self._joint(next_obs)
            joint_next_actions = self._joint(next_raw)

        critic_loss = 0.0
        for aid in self.agent_ids:
            with torch.no_grad():
                q1 = self.critic_targets[aid](
                    self.critic_targets[aid].preprocess(joint_next_obs), joint_next_actions
                )
                q2 = self.critic_2_targets[aid](
                    self.critic_2_targets[aid].preprocess(joint_next_obs), joint_next_actions
                )
                y = rewards[aid].reshape(-1, 1) + self.gamma * (
                    1.0 - dones[aid].reshape(-1, 1)
                ) * torch.minimum(q1, q2)
            qa = self.critics[aid](self.critics[aid].preprocess(joint_obs), joint_actions)
            qb = self.critics_2[aid](self.critics_2[aid].preprocess(joint_obs), joint_actions)
            critic_loss = critic_loss + F.mse_loss(qa, y) + F.mse_loss(qb, y)
        self.critic_optimizer.zero_grad()
        self.critic_2_optimizer.zero_grad()
        critic_loss.backward()
        self.critic_optimizer.step()
        self.critic_2_optimizer.step()

        self._learn_counter += 1
        if self._learn_counter % self.policy_freq == 0:
            actor_loss = 0.0
            current_raw = {}
            for aid in self.agent_ids:
                actor = self.actors[aid]
                actor.train()
                current_raw[aid] = actor(actor.preprocess(obs[aid]))
            for aid in self.agent_ids:
                cols = [
                    current_raw[a] if a == aid else actions[a].reshape(actions[a].shape[0], -1)
                    for a in self.agent_ids
                ]
                joint_a = torch.cat([c.reshape(c.shape[0], -1) for c in cols], dim=1)
                q = self.critics[aid](self.critics[aid].preprocess(joint_obs), joint_a)
                actor_loss = actor_loss + (-q.mean())
            self.actor_optimizer.zero_grad()
            actor_loss.backward()
            self.actor_optimizer.step()
            self.soft_update()
        return float(critic_loss.detach()) / (2 * self.n_agents)

    def soft_update(self) -> None:
        super().soft_update()
        ops.polyak_update_(
            list(self.critic_2_targets.parameters()),
            list(self.critics_2.parameters()),
            self.tau,
        )
