"""MADDPG — multi-agent DDPG with centralized critics.

Reference parity: ``agilerl/algorithms/maddpg.py:61`` — per-agent actors,
centralized critics over concat(all obs, all actions) (``learn`` :694,
stacked actions :722-728), Gumbel-Softmax for discrete action spaces.

MI355X notes: per-agent networks live in :class:`ModuleDict`-s so
architecture mutations broadcast consistently; the per-agent critic/actor
losses are summed into ONE backward pass each (one HIP launch sequence
instead of n_agents sequential graphs); polyak runs as a single fused
kernel over every target parameter.
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional, Tuple

import numpy as np
import torch
import torch.nn.functional as F

from .. import ops
from ..modules.base import ModuleDict
from ..networks.actors import DeterministicActor
from ..networks.q_networks import ContinuousQNetwork
from ..spaces import Box, Discrete, Space, flatdim
from .core.base import MultiAgentRLAlgorithm
from .core.optimizer_wrapper import OptimizerWrapper
from .core.registry import HyperparameterConfig, NetworkGroup, OptimizerConfig, RLParameter

__all__ = ["MADDPG"]


def default_hp_config() -> HyperparameterConfig:
    return HyperparameterConfig(
        lr_actor=RLParameter(min=1e-5, max=1e-2),
        lr_critic=RLParameter(min=1e-5, max=1e-2),
        batch_size=RLParameter(min=16, max=1024, dtype=int),
        learn_step=RLParameter(min=1, max=16, dtype=int),
    )


class MADDPG(MultiAgentRLAlgorithm):
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
        O_U_noise: bool = True,
        expl_noise: float = 0.1,
        mean_noise: float = 0.0,
        theta: float = 0.15,
        dt: float = 1e-2,
        vect_noise_dim: int = 1,
        latent_dim: int = 64,
        shared_agent_groups: Optional[List[List[str]]] = None,
        actor_networks: Optional[Dict[str, Any]] = None,
        critic_networks: Optional[Dict[str, Any]] = None,
        device: str = "cpu",
        **kwargs,
    ):
        super().__init__(
            observation_spaces, action_spaces, agent_ids=agent_ids, index=index,
            learn_step=learn_step, device=device,
            hp_config=hp_config or default_hp_config(), name="MADDPG",
        )
        self._accept_compat_kwargs(**kwargs)
        self.shared_agent_groups = shared_agent_groups
        self.batch_size = int(batch_size)
        self.lr_actor = float(lr_actor)
        self.lr_critic = float(lr_critic)
        self.gamma = float(gamma)
        self.tau = float(tau)
        self.O_U_noise = bool(O_U_noise)
        self.expl_noise = float(expl_noise)
        self.mean_noise = float(mean_noise)
        self.theta = float(theta)
        self.dt = float(dt)
        self.vect_noise_dim = int(vect_noise_dim)
        # per-agent Ornstein-Uhlenbeck state (reference maddpg.py:134 O_U_noise)
        self._ou_state: Dict[str, torch.Tensor] = {}
        self.net_config = net_config
        self.latent_dim = latent_dim

        # joint (centralized) spaces for the critics
        self.joint_obs_dim = sum(flatdim(sp) for sp in self.observation_spaces.values())
        self.joint_action_dim = sum(self._raw_action_dim(sp) for sp in self.action_spaces.values())
        joint_space = Box(-np.inf, np.inf, (self.joint_obs_dim,))

        # grouped-agent net sharing (reference base.py:2330): agents inside a
        # group share ONE actor module object (spaces must match)
        group_of = {}
        for group in shared_agent_groups or []:
            for aid in group:
                group_of[aid] = group[0]
        actor_modules: Dict[str, DeterministicActor] = {}
        actors = {}
        for aid in self.agent_ids:
            leader = group_of.get(aid, aid)
            if leader not in actor_modules:
                if actor_networks is not None and leader in actor_networks:
                    # user-supplied per-agent policy net (reference
                    # maddpg actor_networks constructor arg)
                    from ..networks.base import CustomNetworkAdapter

                    actor_modules[leader] = CustomNetworkAdapter(
                        actor_networks[leader], self.observation_spaces[leader],
                        action_space=self.action_spaces[leader], device=device,
                    )
                else:
                    actor_modules[leader] = DeterministicActor(
                        self.observation_spaces[leader], self.action_spaces[leader],
                        encoder_config=net_config, head_config=head_config,
                        latent_dim=latent_dim, device=device,
                    )
            actors[aid] = actor_modules[leader]
        self.actors = ModuleDict(actors, device=device)
        self.actor_targets = self.actors.clone()
        def _make_critic(aid):
            if critic_networks is not None and aid in critic_networks:
                # user-supplied centralized critic: (joint obs, joint action)
                # -> Q (reference maddpg.py critic_networks)
                from ..networks.base import CustomQAdapter

                return CustomQAdapter(
                    critic_networks[aid], joint_space,
                    action_space=Box(-1.0, 1.0, (self.joint_action_dim,)),
                    device=device,
                )
            return ContinuousQNetwork(
                joint_space, Box(-1.0, 1.0, (self.joint_action_dim,)),
                encoder_config=net_config, head_config=head_config,
                latent_dim=latent_dim, action_dim=self.joint_action_dim, device=device,
            )

        self.critics = ModuleDict(
            {aid: _make_critic(aid) for aid in self.agent_ids},
            device=device,
        )
        self.critic_targets = self.critics.clone()
        for net in (self.actor_targets, self.critic_targets):
            for p in net.parameters():
                p.requires_grad = False

        self.actor_optimizer = OptimizerWrapper(
            torch.optim.Adam, [self.actors], lr=self.lr_actor, multiagent=True
        )
        self.critic_optimizer = OptimizerWrapper(
            torch.optim.Adam, [self.critics], lr=self.lr_critic, multiagent=True
        )

        self.register_network_group(
            NetworkGroup(eval_network="actors", shared_networks=["actor_targets"], policy=True, multiagent=True)
        )
        self.register_network_group(
            NetworkGroup(eval_network="critics", shared_networks=["critic_targets"], multiagent=True)
        )
        self.register_optimizer(
            OptimizerConfig(name="actor_optimizer", networks=["actors"], lr_name="lr_actor")
        )
        self.register_optimizer(
            OptimizerConfig(name="critic_optimizer", networks=["critics"], lr_name="lr_critic")
        )
        self.register_mutation_hook("_sync_targets_after_mutation")

    # ------------------------------------------------------------------
    @staticmethod
    def _raw_action_dim(space: Space) -> int:
        return space.n if isinstance(space, Discrete) else flatdim(space)

    def _sync_targets_after_mutation(self) -> None:
        self.actor_targets.load_state_dict(self.actors.state_dict())
        self.critic_targets.load_state_dict(self.critics.state_dict())
        for net in (self.actor_targets, self.critic_targets):
            for p in net.parameters():
                p.requires_grad = False

    # ------------------------------------------------------------------
    def _sample_noise(self, aid: str, like: torch.Tensor) -> torch.Tensor:
        """OU (stateful, mean-reverting) or Gaussian exploration noise for one
        agent.  Reference maddpg.py O_U_noise/mean_noise/theta/dt."""
        if not self.O_U_noise:
            return torch.randn_like(like) * self.expl_noise + self.mean_noise
        state = self._ou_state.get(aid)
        if state is None or state.shape != like.shape or state.device != like.device:
            state = torch.zeros_like(like)
        dx = self.theta * (self.mean_noise - state) * self.dt + (
            self.expl_noise * (self.dt ** 0.5)
        ) * torch.randn_like(like)
        state = state + dx
        self._ou_state[aid] = state
        return state

    def reset_action_noise(self, indices=None) -> None:
        """Reset OU state (per finished env row, or all)."""
        for state in self._ou_state.values():
            if indices is None:
                state.zero_()
            else:
                state[np.asarray(indices)] = 0.0

    def get_action(
        self, obs: Dict[str, np.ndarray], training: bool = True, **kwargs
    ) -> Tuple[Dict[str, np.ndarray], Dict[str, np.ndarray]]:
        """Returns (env_actions, raw_actions).  Raw actions are what the
        centralized critics consume (one-hot / continuous vectors)."""
        first = obs[self.agent_ids[0]]
        device_native = isinstance(first, torch.Tensor) and first.is_cuda
        env_actions, raw_actions = {}, {}
        with torch.no_grad():
            for aid in self.agent_ids:
                actor = self.actors[aid]
                actor.train(training)
                out = actor(actor.preprocess(obs[aid]))
                space = self.action_spaces[aid]
                if isinstance(space, Discrete):
                    raw = out  # gumbel-softmax one-hot (train) / hard one-hot (eval)
                    act = raw.argmax(-1)
                    env_actions[aid] = act if device_native else act.cpu().numpy()
                    raw_actions[aid] = raw if device_native else raw.cpu().numpy()
                else:
                    a = out
                    if training and (self.expl_noise > 0 or self.mean_noise != 0):
                        a = a + self._sample_noise(aid, a)
                    low = torch.as_tensor(space.low, device=a.device, dtype=a.dtype)
                    high = torch.as_tensor(space.high, device=a.device, dtype=a.dtype)
                    a = a.clamp(low.min(), high.max())
                    env_actions[aid] = a if device_native else a.cpu().numpy()
                    raw_actions[aid] = env_actions[aid]
        return env_actions, raw_actions

    # ------------------------------------------------------------------
    def _joint(self, d: Dict[str, torch.Tensor]) -> torch.Tensor:
        return torch.cat([d[aid].reshape(d[aid].shape[0], -1) for aid in self.agent_ids], dim=1)

    def _to_dev(self, d) -> Dict[str, torch.Tensor]:
        out = {}
        for aid, v in d.items():
            t = v if isinstance(v, torch.Tensor) else torch.as_tensor(np.asarray(v))
            out[aid] = t.float().to(self.device)
        return out

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

        # --- critics: one summed backward over all agents
        critic_loss = 0.0
        for aid in self.agent_ids:
            with torch.no_grad():
                q_next = self.critic_targets[aid](
                    self.critic_targets[aid].preprocess(joint_next_obs), joint_next_actions
                )
                y = rewards[aid].reshape(-1, 1) + self.gamma * (
                    1.0 - dones[aid].reshape(-1, 1)
                ) * q_next
            q = self.critics[aid](self.critics[aid].preprocess(joint_obs), joint_actions)
            critic_loss = critic_loss + F.mse_loss(q, y)
        self.critic_optimizer.zero_grad()
        critic_loss.backward()
        self.critic_optimizer.step()

        # --- actors: each agent's action column replaced by its live actor
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
        return float(critic_loss.detach()) / self.n_agents

    def soft_update(self) -> None:
        ops.polyak_update_(
            list(self.actor_targets.parameters()) + list(self.critic_targets.parameters()),
            list(self.actors.parameters()) + list(self.critics.parameters()),
            self.tau,
        )

    # ------------------------------------------------------------------
    def test(self, env, max_steps: Optional[int] = None, loop: int = 3, **kwargs) -> float:
        with torch.no_grad():
            totals = []
            for _ in range(loop):
                obs, _ = env.reset()
                ep_rew = np.zeros(env.num_envs)
                steps = 0
                while True:
                    env_actions, _ = self.get_action(obs, training=False)
                    obs, rewards, term, trunc, _ = env.step(env_actions)
                    ep_rew += np.mean([rewards[a] for a in self.agent_ids], axis=0)
                    steps += 1
                    done = np.any([term[a] | trunc[a] for a in self.agent_ids], axis=0)
                    if done.all() or (max_steps is not None and steps >= max_steps):
                        break
                totals.append(ep_rew.mean())
        fitness = float(np.mean(totals))
        self.fitness.append(fitness)
        return fitness
