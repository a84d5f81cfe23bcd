"""NeuralTS — neural Thompson-sampling contextual bandit.

Reference parity: ``agilerl/algorithms/neural_ts_bandit.py:33``.  Same
confidence sketch as NeuralUCB but samples the score from
``N(mu, nu^2 * g^T Sigma^-1 g)`` instead of adding the bonus.
"""

from __future__ import annotations

import numpy as np
import torch

from typing import Any, Dict, Optional

from ..spaces import Discrete, Space
from .core.registry import HyperparameterConfig
from .neural_ucb import NeuralUCB

__all__ = ["NeuralTS"]


class NeuralTS(NeuralUCB):
    # explicit signature (not *args) so AlgorithmMeta captures init args —
    # clone()/checkpoint/evolution rebuild from them
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
        device: str = "cpu",
        **kwargs,
    ):
        super().__init__(
            observation_space, action_space, index=index, hp_config=hp_config,
            net_config=net_config, head_config=head_config, batch_size=batch_size,
            lr=lr, learn_step=learn_step, gamma=gamma, lamb=lamb, reg=reg,
            latent_dim=latent_dim, device=device,
            **kwargs,
        )
        self.algo = "NeuralTS"

    def get_action(self, context, training: bool = True, **kwargs) -> int:
        ctx = torch.as_tensor(np.asarray(context), dtype=torch.float32, device=self.device)
        scores = []
        for k in range(ctx.shape[0]):
            mu = self.actor(self.actor.preprocess(ctx[k : k + 1])).squeeze()
            if training:
                g = self._grad_vector(mu)
                sigma = float(self._exploration_bonus(g))
                sample = float(np.random.normal(float(mu.detach()), max(sigma, 1e-8)))
                scores.append((sample, g))
            else:
                scores.append((float(mu.detach()), None))
        arm = int(np.argmax([s[0] for s in scores]))
        if training and scores[arm][1] is not None:
            g = scores[arm][1]
            self.sigma_inv = 1.0 / (1.0 / self.sigma_inv + g * g)
        return arm
