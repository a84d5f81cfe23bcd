"""NeuralUCB contextual bandit.

Reference parity: ``agilerl/algorithms/neural_ucb_bandit.py:33`` — neural
reward model with gradient-based UCB confidence (diagonal sketch of the
design matrix), mutation-aware confidence-state reinit
(reference mutation.py:1196).
"""

from __future__ import annotations

from typing import Any, Dict, Optional

import numpy as np
import torch
import torch.nn.functional as F

from ..networks.base import EvolvableNetwork
from ..spaces import Discrete, Space
from .core.base import RLAlgorithm
from .core.optimizer_wrapper import OptimizerWrapper
from .core.registry import HyperparameterConfig, NetworkGroup, OptimizerConfig, RLParameter

__all__ = ["NeuralUCB"]


def default_hp_config() -> HyperparameterConfig:
    return HyperparameterConfig(
        lr=RLParameter(min=1e-4, max=1e-1),
        batch_size=RLParameter(min=16, max=512, dtype=int),
        gamma=RLParameter(min=0.1, max=10.0),
    )


class NeuralUCB(RLAlgorithm):
    def __init__(
        self,
        observation_space: Space,
        action_space: Discrete,
        index: int = 0,
        hp_config: Optional[HyperparameterConfig] = None,
        net_config: Optional[Dict[str, Any]] = None,
        head_config: Optional[Dict[str, Any]] = None,
        batch_size: int = 64,
        lr: float = 1e-3,
        learn_step: int = 2,
        gamma: float = 1.0,
        lamb: float = 1.0,
        reg: float = 0.000625,
        latent_dim: int = 64,
        actor_network=None,
        device: str = "cpu",
        **kwargs,
    ):
        super().__init__(
            observation_space, action_space, index=index, learn_step=learn_step,
            device=device, hp_config=hp_config or default_hp_config(), name="NeuralUCB",
        )
        self._accept_compat_kwargs(**kwargs)
        self.batch_size = int(batch_size)
        self.lr = float(lr)
        self.gamma = float(gamma)
        self.lamb = float(lamb)
        self.reg = float(reg)
        self.net_config = net_config
        self.latent_dim = latent_dim

        if actor_network is not None:
            # user-supplied scorer: preprocessed context -> (B,1) reward
            # estimate (reference neural_ucb_bandit.py actor_network)
            from ..networks.base import CustomNetworkAdapter

            self.actor = CustomNetworkAdapter(actor_network, observation_space,
                                              device=device)
        else:
            self.actor = EvolvableNetwork(
                observation_space, num_outputs=1, encoder_config=net_config,
                head_config=head_config, latent_dim=latent_dim, device=device,
            )
        self.optimizer = OptimizerWrapper(torch.optim.Adam, [self.actor], lr=self.lr)
        self.register_network_group(NetworkGroup(eval_network="actor", policy=True))
        self.register_optimizer(OptimizerConfig(name="optimizer", networks=["actor"], lr_name="lr"))
        self.register_mutation_hook("reinit_confidence")
        self.reinit_confidence()

    # ------------------------------------------------------------------
    def reinit_confidence(self) -> None:
        """(Re)build the diagonal design-matrix sketch after any mutation."""
        self.numel = sum(p.numel() for p in self.actor.parameters() if p.requires_grad)
        self.sigma_inv = torch.full((self.numel,), 1.0 / self.lamb, device=self.device)

    def _grad_vector(self, score: torch.Tensor) -> torch.Tensor:
        grads = torch.autograd.grad(
            score, [p for p in self.actor.parameters() if p.requires_grad],
            retain_graph=False, create_graph=False, allow_unused=True,
        )
        flat = [
            (g if g is not None else torch.zeros_like(p)).reshape(-1)
            for g, p in zip(grads, (p for p in self.actor.parameters() if p.requires_grad))
        ]
        return torch.cat(flat)

    def _exploration_bonus(self, g: torch.Tensor) -> torch.Tensor:
        return self.gamma * torch.sqrt((g * g * self.sigma_inv).sum())

    def get_action(self, context, training: bool = True, **kwargs) -> int:
        """context: (num_arms, context_dim) -> chosen arm index."""
        ctx = torch.as_tensor(np.asarray(context), dtype=torch.float32, device=self.device)
        scores = []
        chosen_g = None
        for k in range(ctx.shape[0]):
            mu = self.actor(self.actor.preprocess(ctx[k : k + 1])).squeeze()
            if training:
                g = self._grad_vector(mu)
                ucb = mu.detach() + self._exploration_bonus(g)
                scores.append((float(ucb), g))
            else:
                scores.append((float(mu.detach()), None))
        arm = int(np.argmax([s[0] for s in scores]))
        if training and scores[arm][1] is not None:
            g = scores[arm][1]
            self.sigma_inv = 1.0 / (1.0 / self.sigma_inv + g * g)
        return arm

    # ------------------------------------------------------------------
    def learn(self, experiences: Dict[str, torch.Tensor]) -> float:
        obs = experiences["obs"]
        rewards = experiences["reward"].to(self.device).float().reshape(-1, 1)
        pred = self.actor(self.actor.preprocess(obs))
        loss = F.mse_loss(pred, rewards)
        reg_loss = self.reg * sum((p**2).sum() for p in self.actor.parameters())
        total = loss + reg_loss
        self.optimizer.zero_grad()
        total.backward()
        self.optimizer.step()
        return float(loss.detach())

    # ------------------------------------------------------------------
    def test(self, env, max_steps: Optional[int] = 200, loop: int = 1, **kwargs) -> float:
        rewards = []
        with torch.no_grad():
            pass
        for _ in range(loop):
            context = env.reset()
            total = 0.0
            for _ in range(max_steps or 200):
                arm = self.get_action(context, training=False)
                r, context = env.step(arm)
                total += r
            rewards.append(total / (max_steps or 200))
        fitness = float(np.mean(rewards))
        self.fitness.append(fitness)
        return fitness
