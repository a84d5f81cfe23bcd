"""IPPO — independent PPO per agent (multi-agent on-policy).

Reference parity: ``agilerl/algorithms/ippo.py:59``.
Each agent id owns an independent StochasticActor + ValueNetwork (in
ModuleDicts so mutations stay structurally consistent); the learn step
sums all agents' clip losses into one backward.
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional

import numpy as np
import torch
import torch.nn as nn

from ..components.rollout_buffer import RolloutBuffer
from ..modules.base import ModuleDict
from ..networks.actors import StochasticActor
from ..networks.value_networks import ValueNetwork
from ..spaces import Space
from .core.base import MultiAgentRLAlgorithm
from .core.optimizer_wrapper import OptimizerWrapper
from .core.registry import HyperparameterConfig, NetworkGroup, OptimizerConfig, RLParameter

__all__ = ["IPPO"]


def default_hp_config() -> HyperparameterConfig:
    return HyperparameterConfig(
        lr=RLParameter(min=1e-5, max=1e-2),
        batch_size=RLParameter(min=64, max=4096, dtype=int),
        clip_coef=RLParameter(min=0.05, max=0.4),
        ent_coef=RLParameter(min=1e-4, max=0.05),
    )


class IPPO(MultiAgentRLAlgorithm):
    def __init__(
        self,
        observation_spaces: Dict[str, Space],
        action_spaces: Dict[str, Space],
        agent_ids: Optional[List[str]] = None,
        index: int = 0,
        hp_config: Optional[HyperparameterConfig] = None,
        net_config: Optional[Dict[str, Any]] = None,
        head_config: Optional[Dict[str, Any]] = None,
        batch_size: int = 512,
        lr: float = 3e-4,
        learn_step: int = 128,
        gamma: float = 0.99,
        gae_lambda: float = 0.95,
        clip_coef: float = 0.2,
        ent_coef: float = 0.01,
        vf_coef: float = 0.5,
        update_epochs: int = 4,
        max_grad_norm: float = 0.5,
        target_kl: Optional[float] = None,
        action_std_init: Optional[float] = None,
        action_batch_size: Optional[int] = None,
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
            hp_config=hp_config or default_hp_config(), name="IPPO",
        )
        self._accept_compat_kwargs(**kwargs)
        self.batch_size = int(batch_size)
        self.lr = float(lr)
        self.gamma = float(gamma)
        self.gae_lambda = float(gae_lambda)
        self.clip_coef = float(clip_coef)
        self.ent_coef = float(ent_coef)
        self.vf_coef = float(vf_coef)
        self.update_epochs = int(update_epochs)
        self.max_grad_norm = float(max_grad_norm)
        # reference ippo.py: target_kl early-stop, action_std_init (initial
        # LOG std, ppo.py:279 semantics), action_batch_size (chunked
        # get_action for very wide vector envs)
        self.target_kl = target_kl
        self.action_std_init = action_std_init
        self.action_batch_size = action_batch_size
        self.net_config = net_config
        self.latent_dim = latent_dim
        self.shared_agent_groups = shared_agent_groups

        # homogeneous agent groups share one actor/critic (reference
        # grouped-agent sharing, base.py:2330)
        group_of = {}
        for group in shared_agent_groups or []:
            for aid in group:
                group_of[aid] = group[0]
        actor_mods, critic_mods, actors, critics = {}, {}, {}, {}
        for aid in self.agent_ids:
            leader = group_of.get(aid, aid)
            if leader not in actor_mods:
                if actor_networks is not None and leader in actor_networks:
                    from ..networks.base import CustomStochasticAdapter

                    actor_mods[leader] = CustomStochasticAdapter(
                        actor_networks[leader], self.observation_spaces[leader],
                        self.action_spaces[leader], device=device,
                    )
                else:
                    actor_mods[leader] = StochasticActor(
                        self.observation_spaces[leader], self.action_spaces[leader],
                        encoder_config=net_config, head_config=head_config,
                        latent_dim=latent_dim, device=device,
                        log_std_init=float(action_std_init or 0.0),
                    )
                if critic_networks is not None and leader in critic_networks:
                    from ..networks.base import CustomNetworkAdapter

                    critic_mods[leader] = CustomNetworkAdapter(
                        critic_networks[leader], self.observation_spaces[leader],
                        device=device,
                    )
                else:
                    critic_mods[leader] = ValueNetwork(
                        self.observation_spaces[leader], encoder_config=net_config,
                        head_config=head_config, latent_dim=latent_dim, device=device,
                    )
            actors[aid] = actor_mods[leader]
            critics[aid] = critic_mods[leader]
        self.actors = ModuleDict(actors, device=device)
        self.critics = ModuleDict(critics, device=device)
        self.optimizer = OptimizerWrapper(
            torch.optim.Adam, [self.actors, self.critics], lr=self.lr
        )
        self.register_network_group(NetworkGroup(eval_network="actors", policy=True, multiagent=True))
        self.register_network_group(NetworkGroup(eval_network="critics", multiagent=True))
        self.register_optimizer(
            OptimizerConfig(name="optimizer", networks=["actors", "critics"], lr_name="lr")
        )

    # ------------------------------------------------------------------
    def get_action(self, obs: Dict[str, np.ndarray], training: bool = True, **kwargs):
        """Training: (env_actions, log_probs, values) dicts; eval: env_actions."""
        env_actions, log_probs, values = {}, {}, {}
        with torch.no_grad():
            for aid in self.agent_ids:
                actor = self.actors[aid]
                pre = actor.preprocess(obs[aid])
                if not training:
                    env_actions[aid] = actor.deterministic_action(pre).cpu().numpy()
                    continue
                abs_ = self.action_batch_size
                if abs_ is not None and pre.shape[0] > abs_:
                    # reference action_batch_size: chunk very wide vector
                    # envs through the actor to bound activation memory
                    outs = [actor.sample(pre[i:i + abs_]) for i in range(0, pre.shape[0], abs_)]
                    a = torch.cat([o[0] for o in outs])
                    lp = torch.cat([o[1] for o in outs])
                else:
                    a, lp, _ = actor.sample(pre)
                env_actions[aid] = a.cpu().numpy()
                log_probs[aid] = lp
                values[aid] = self.critics[aid](self.critics[aid].preprocess(obs[aid])).squeeze(-1)
        if not training:
            return env_actions
        return env_actions, log_probs, values

    def get_values(self, obs: Dict[str, np.ndarray]) -> Dict[str, torch.Tensor]:
        with torch.no_grad():
            return {
                aid: self.critics[aid](self.critics[aid].preprocess(obs[aid])).squeeze(-1)
                for aid in self.agent_ids
            }

    # ------------------------------------------------------------------
    def learn(self, buffers: Dict[str, RolloutBuffer]) -> Dict[str, float]:
        stats = {"policy_loss": 0.0, "value_loss": 0.0, "entropy": 0.0, "approx_kl": 0.0}
        n = 0
        for _ in range(self.update_epochs):
            epoch_kl, epoch_mbs = 0.0, 0
            iters = {aid: buffers[aid].get_minibatches(self.batch_size) for aid in self.agent_ids}
            while True:
                mbs = {}
                for aid, it in iters.items():
                    mb = next(it, None)
                    if mb is not None:
                        mbs[aid] = mb
                if not mbs:
                    break
                loss = 0.0
                for aid, mb in mbs.items():
                    actor, critic = self.actors[aid], self.critics[aid]
                    adv = mb["advantages"].reshape(-1)
                    if adv.numel() > 1:
                        adv = (adv - adv.mean()) / (adv.std() + 1e-8)
                    lp, ent = actor.evaluate_actions(
                        actor.preprocess(mb["obs"]), mb["action"].to(self.device)
                    )
                    ratio = (lp.reshape(-1) - mb["log_prob"].reshape(-1)).exp()
                    pg = torch.maximum(
                        -adv * ratio, -adv * ratio.clamp(1 - self.clip_coef, 1 + self.clip_coef)
                    ).mean()
                    v = critic(critic.preprocess(mb["obs"])).reshape(-1)
                    vloss = 0.5 * ((v - mb["returns"].reshape(-1)) ** 2).mean()
                    loss = loss + pg + self.vf_coef * vloss - self.ent_coef * ent.mean()
                    with torch.no_grad():
                        kl = float((ratio - 1 - ratio.log()).mean())
                        stats["approx_kl"] += kl
                        epoch_kl += kl
                        epoch_mbs += 1
                    stats["policy_loss"] += float(pg.detach())
                    stats["value_loss"] += float(vloss.detach())
                    stats["entropy"] += float(ent.mean().detach())
                self.optimizer.zero_grad()
                loss.backward()
                nn.utils.clip_grad_norm_(
                    list(self.actors.parameters()) + list(self.critics.parameters()),
                    self.max_grad_norm,
                )
                self.optimizer.step()
                n += len(mbs)
            if (
                self.target_kl is not None
                and epoch_mbs
                and epoch_kl / epoch_mbs > self.target_kl
            ):
                break  # reference PPO-family target_kl early stop
        if n:
            stats = {k: v / n for k, v in stats.items()}
        return stats

    # ------------------------------------------------------------------
    def test(self, env, max_steps: Optional[int] = None, loop: int = 3, **kwargs) -> float:
        with torch.no_grad():
            totals = []
            for _ in range(loop):
                obs, _ = env.reset()
                ep_rew = np.zeros(env.num_envs)
                steps = 0
                while True:
                    env_actions = self.get_action(obs, training=False)
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
