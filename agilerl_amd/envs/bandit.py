"""Contextual bandit environments.

Reference parity: ``agilerl/wrappers/learning.py:66`` (BanditEnv: labelled
dataset -> contextual bandit) plus a synthetic generator for tests/benchmarks.

API: ``context = env.reset()`` -> (num_arms, context_dim);
``reward, next_context = env.step(arm)``; regret tracking built in.
"""

from __future__ import annotations

from typing import Optional, Tuple

import numpy as np

from ..spaces import Box, Discrete

__all__ = ["SyntheticBanditEnv", "BanditEnv"]


class SyntheticBanditEnv:
    """Nonlinear synthetic bandit: reward_k = f(w_k . x) + noise."""

    def __init__(
        self,
        context_dim: int = 8,
        num_arms: int = 4,
        noise: float = 0.05,
        seed: Optional[int] = None,
    ):
        self.context_dim = context_dim
        self.num_arms = num_arms
        self.arms = num_arms
        self.noise = noise
        self.rng = np.random.default_rng(seed)
        self.w = self.rng.normal(size=(num_arms, context_dim)) / np.sqrt(context_dim)
        self.observation_space = Box(-np.inf, np.inf, (context_dim,))
        self.action_space = Discrete(num_arms)
        self.regret: list = []
        self._context: Optional[np.ndarray] = None

    def _expected(self, context: np.ndarray) -> np.ndarray:
        return np.cos(3 * (self.w * context).sum(axis=1)) ** 2

    def _new_context(self) -> np.ndarray:
        x = self.rng.normal(size=(self.context_dim,))
        x /= np.linalg.norm(x) + 1e-8
        self._context = np.tile(x, (self.num_arms, 1)).astype(np.float32)
        return self._context

    def reset(self) -> np.ndarray:
        self.regret = []
        return self._new_context()

    def step(self, arm: int) -> Tuple[float, np.ndarray]:
        mu = self._expected(self._context)
        reward = float(mu[int(arm)] + self.rng.normal(0, self.noise))
        self.regret.append(float(mu.max() - mu[int(arm)]))
        return reward, self._new_context()


class BanditEnv:
    """Labelled-dataset bandit: features (N, F), labels (N,); pulling the
    arm equal to the label yields reward 1 (reference wrappers/learning.py:66)."""

    def __init__(self, features: np.ndarray, targets: np.ndarray, seed: Optional[int] = None):
        self.features = np.asarray(features, dtype=np.float32)
        targets = np.asarray(targets).reshape(-1)
        classes = np.unique(targets)
        self.class_map = {c: i for i, c in enumerate(classes)}
        self.targets = np.array([self.class_map[c] for c in targets])
        self.num_arms = self.arms = len(classes)
        self.context_dim = self.features.shape[1] * self.num_arms
        self.observation_space = Box(-np.inf, np.inf, (self.context_dim,))
        self.action_space = Discrete(self.num_arms)
        self.rng = np.random.default_rng(seed)
        self.regret: list = []
        self._idx = 0

    def _context_for(self, idx: int) -> np.ndarray:
        """Arm k's context: the feature row placed in block k (disjoint-arm
        encoding)."""
        f = self.features[idx]
        ctx = np.zeros((self.num_arms, self.context_dim), dtype=np.float32)
        F = self.features.shape[1]
        for k in range(self.num_arms):
            ctx[k, k * F : (k + 1) * F] = f
        return ctx

    def reset(self) -> np.ndarray:
        self.regret = []
        self._idx = int(self.rng.integers(len(self.features)))
        return self._context_for(self._idx)

    def step(self, arm: int) -> Tuple[float, np.ndarray]:
        reward = 1.0 if int(arm) == int(self.targets[self._idx]) else 0.0
        self.regret.append(1.0 - reward)
        self._idx = int(self.rng.integers(len(self.features)))
        return reward, self._context_for(self._idx)
