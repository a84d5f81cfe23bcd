"""CQN — conservative Q-learning for offline discrete control.

Reference parity: ``agilerl/algorithms/cqn.py:40``.  DQN backbone with a
CQL regularizer: ``alpha * (logsumexp_a Q(s,a) - Q(s, a_data))`` keeps
out-of-distribution action values pessimistic.
"""

from __future__ import annotations

from typing import Any, Dict, Optional

import torch
import torch.nn.functional as F

from ..spaces import Space
from .core.registry import HyperparameterConfig
from .dqn import DQN

__all__ = ["CQN"]


class CQN(DQN):
    def __init__(
        self,
        observation_space: Space,
        action_space: Space,
        index: int = 0,
        hp_config: Optional[HyperparameterConfig] = None,
        net_config: Optional[Dict[str, Any]] = None,
        head_config: Optional[Dict[str, Any]] = None,
        batch_size: int = 64,
        lr: float = 1e-4,
        learn_step: int = 5,
        gamma: float = 0.99,
        tau: float = 1e-3,
        double: bool = True,
        cql_alpha: float = 1.0,
        latent_dim: int = 64,
        actor_network=None,
        device: str = "cpu",
        **kwargs,
    ):
        super().__init__(
            observation_space, action_space, index=index, hp_config=hp_config,
            net_config=net_config, head_config=head_config, batch_size=batch_size,
            lr=lr, learn_step=learn_step, gamma=gamma, tau=tau, double=double,
            latent_dim=latent_dim, actor_network=actor_network, device=device,
            **kwargs,
        )
        self.algo = "CQN"
        self.cql_alpha = float(cql_alpha)

    def learn(self, experiences: Dict[str, torch.Tensor]) -> float:
        obs = experiences["obs"]
        actions = experiences["action"].to(self.device).long().reshape(-1)
        rewards = experiences["reward"].to(self.device).float().reshape(-1)
        next_obs = experiences["next_obs"]
        dones = experiences["done"].to(self.device).float().reshape(-1)

        with torch.no_grad():
            q_next_t = self.actor_target(self.actor_target.preprocess(next_obs))
            if self.double:
                sel = self.actor(self.actor.preprocess(next_obs)).argmax(-1, keepdim=True)
                q_next = q_next_t.gather(1, sel).squeeze(-1)
            else:
                q_next = q_next_t.max(dim=-1).values
            target = rewards + (1.0 - dones) * self.gamma * q_next

        q_all = self.actor(self.actor.preprocess(obs))
        q_pred = q_all.gather(1, actions.unsqueeze(1)).squeeze(-1)
        td_loss = F.huber_loss(q_pred, target)
        cql_penalty = (torch.logsumexp(q_all, dim=-1) - q_pred).mean()
        loss = td_loss + self.cql_alpha * cql_penalty

        self.optimizer.zero_grad()
        loss.backward()
        self.optimizer.step()
        self.soft_update()
        return float(loss.detach())
