"""Analytic probe environments for learning-correctness tests.

Reference parity: ``agilerl/utils/probe_envs.py`` (ConstantRewardEnv :29 …
PolicyContActionsEnv :966).  Each env has known correct Q / V / policy
values; tests train briefly and assert convergence.
All are vectorized (BatchedVecEnv) with episode length 1 or 2.
"""

from __future__ import annotations

from typing import Optional

import numpy as np

from ..spaces import Box, Discrete
from .base import BatchedVecEnv

__all__ = [
    "ConstantRewardEnv",
    "ObsDependentRewardEnv",
    "DiscountedRewardEnv",
    "FixedObsPolicyEnv",
    "PolicyEnv",
    "ConstantRewardContActionsEnv",
    "FixedObsPolicyContActionsEnv",
]


class ConstantRewardEnv(BatchedVecEnv):
    """Always reward 1, one step.  Q*(s, a) = 1 for all a."""

    max_episode_steps = 1
    q_values = np.array([[1.0, 1.0]])
    v_values = np.array([[1.0]])

    def __init__(self, num_envs: int = 1, seed: Optional[int] = None):
        super().__init__(num_envs, seed)
        self.single_observation_space = Box(0.0, 1.0, (1,))
        self.single_action_space = Discrete(2)

    def _reset_rows(self, mask):
        pass

    def _obs(self):
        return np.zeros((self.num_envs, 1), dtype=np.float32)

    def _step_all(self, actions):
        return (
            np.ones(self.num_envs, dtype=np.float32),
            np.ones(self.num_envs, dtype=bool),
            None,
        )


class ObsDependentRewardEnv(BatchedVecEnv):
    """Obs 0 -> reward -1, obs 1 -> reward +1, one step.

    Q*(s=0, a)=-1, Q*(s=1, a)=+1.
    """

    max_episode_steps = 1
    q_values = np.array([[-1.0, -1.0], [1.0, 1.0]])
    v_values = np.array([[-1.0], [1.0]])

    def __init__(self, num_envs: int = 1, seed: Optional[int] = None):
        super().__init__(num_envs, seed)
        self.single_observation_space = Box(0.0, 1.0, (1,))
        self.single_action_space = Discrete(2)
        self.state = np.zeros(self.num_envs, dtype=np.float32)

    def _reset_rows(self, mask):
        self.state[mask] = self.rng.integers(0, 2, int(mask.sum())).astype(np.float32)

    def _obs(self):
        return self.state.reshape(-1, 1).copy()

    def _step_all(self, actions):
        reward = np.where(self.state > 0.5, 1.0, -1.0).astype(np.float32)
        return reward, np.ones(self.num_envs, dtype=bool), None


class DiscountedRewardEnv(BatchedVecEnv):
    """Two steps: obs 0 then obs 1; reward 1 only on the second step.

    Q*(s=0) = gamma, Q*(s=1) = 1 — tests the discount/bootstrap path.
    """

    max_episode_steps = 2

    def __init__(self, num_envs: int = 1, seed: Optional[int] = None):
        super().__init__(num_envs, seed)
        self.single_observation_space = Box(0.0, 1.0, (1,))
        self.single_action_space = Discrete(2)
        self.phase = np.zeros(self.num_envs, dtype=np.float32)

    def q_values_for(self, gamma: float) -> np.ndarray:
        return np.array([[gamma, gamma], [1.0, 1.0]])

    def _reset_rows(self, mask):
        self.phase[mask] = 0.0

    def _obs(self):
        return self.phase.reshape(-1, 1).copy()

    def _step_all(self, actions):
        second = self.phase > 0.5
        reward = np.where(second, 1.0, 0.0).astype(np.float32)
        terminated = second.copy()
        self.phase = np.where(second, self.phase, 1.0)
        return reward, terminated, None


class FixedObsPolicyEnv(BatchedVecEnv):
    """Fixed obs; action 0 -> +1, action 1 -> -1.  Optimal policy: action 0."""

    max_episode_steps = 1
    q_values = np.array([[1.0, -1.0]])

    def __init__(self, num_envs: int = 1, seed: Optional[int] = None):
        super().__init__(num_envs, seed)
        self.single_observation_space = Box(0.0, 1.0, (1,))
        self.single_action_space = Discrete(2)

    def _reset_rows(self, mask):
        pass

    def _obs(self):
        return np.zeros((self.num_envs, 1), dtype=np.float32)

    def _step_all(self, actions):
        reward = np.where(actions.reshape(-1) == 0, 1.0, -1.0).astype(np.float32)
        return reward, np.ones(self.num_envs, dtype=bool), None


class PolicyEnv(BatchedVecEnv):
    """One-hot obs in {e_0, e_1}; reward +1 iff action == obs index."""

    max_episode_steps = 1

    def __init__(self, num_envs: int = 1, seed: Optional[int] = None):
        super().__init__(num_envs, seed)
        self.single_observation_space = Box(0.0, 1.0, (2,))
        self.single_action_space = Discrete(2)
        self.idx = np.zeros(self.num_envs, dtype=np.int64)

    def _reset_rows(self, mask):
        self.idx[mask] = self.rng.integers(0, 2, int(mask.sum()))

    def _obs(self):
        obs = np.zeros((self.num_envs, 2), dtype=np.float32)
        obs[np.arange(self.num_envs), self.idx] = 1.0
        return obs

    def _step_all(self, actions):
        reward = np.where(actions.reshape(-1) == self.idx, 1.0, -1.0).astype(np.float32)
        return reward, np.ones(self.num_envs, dtype=bool), None


class ConstantRewardContActionsEnv(BatchedVecEnv):
    """Continuous actions, always reward 1, one step. Q* = 1."""

    max_episode_steps = 1
    q_values = np.array([[1.0]])

    def __init__(self, num_envs: int = 1, seed: Optional[int] = None):
        super().__init__(num_envs, seed)
        self.single_observation_space = Box(0.0, 1.0, (1,))
        self.single_action_space = Box(-1.0, 1.0, (1,))

    def _reset_rows(self, mask):
        pass

    def _obs(self):
        return np.zeros((self.num_envs, 1), dtype=np.float32)

    def _step_all(self, actions):
        return (
            np.ones(self.num_envs, dtype=np.float32),
            np.ones(self.num_envs, dtype=bool),
            None,
        )


class FixedObsPolicyContActionsEnv(BatchedVecEnv):
    """Fixed obs; reward = -(action - 0.5)^2.  Optimal action 0.5."""

    max_episode_steps = 1
    target_action = 0.5

    def __init__(self, num_envs: int = 1, seed: Optional[int] = None):
        super().__init__(num_envs, seed)
        self.single_observation_space = Box(0.0, 1.0, (1,))
        self.single_action_space = Box(-1.0, 1.0, (1,))

    def _reset_rows(self, mask):
        pass

    def _obs(self):
        return np.zeros((self.num_envs, 1), dtype=np.float32)

    def _step_all(self, actions):
        a = np.asarray(actions, dtype=np.float64).reshape(self.num_envs, -1)[:, 0]
        reward = -((a - self.target_action) ** 2)
        return reward.astype(np.float32), np.ones(self.num_envs, dtype=bool), None
