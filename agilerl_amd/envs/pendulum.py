"""Vectorized Pendulum (continuous control, batched numpy dynamics)."""

from __future__ import annotations

from typing import Optional

import numpy as np

from ..spaces import Box
from .base import BatchedVecEnv

__all__ = ["PendulumVecEnv"]


class PendulumVecEnv(BatchedVecEnv):
    max_episode_steps = 200

    MAX_SPEED = 8.0
    MAX_TORQUE = 2.0
    DT = 0.05
    G = 10.0
    M = 1.0
    L = 1.0

    def __init__(self, num_envs: int = 1, seed: Optional[int] = None):
        super().__init__(num_envs, seed)
        high = np.array([1.0, 1.0, self.MAX_SPEED], dtype=np.float32)
        self.single_observation_space = Box(-high, high)
        self.single_action_space = Box(-self.MAX_TORQUE, self.MAX_TORQUE, (1,))
        self.theta = np.zeros(self.num_envs)
        self.theta_dot = np.zeros(self.num_envs)

    def _reset_rows(self, mask):
        n = int(mask.sum())
        self.theta[mask] = self.rng.uniform(-np.pi, np.pi, n)
        self.theta_dot[mask] = self.rng.uniform(-1.0, 1.0, n)

    def _obs(self):
        return np.stack(
            [np.cos(self.theta), np.sin(self.theta), self.theta_dot], axis=1
        ).astype(np.float32)

    def _step_all(self, actions):
        u = np.clip(np.asarray(actions, dtype=np.float64).reshape(self.num_envs, -1)[:, 0],
                    -self.MAX_TORQUE, self.MAX_TORQUE)
        th = ((self.theta + np.pi) % (2 * np.pi)) - np.pi
        cost = th**2 + 0.1 * self.theta_dot**2 + 0.001 * u**2
        new_dot = self.theta_dot + (
            3 * self.G / (2 * self.L) * np.sin(th) + 3.0 / (self.M * self.L**2) * u
        ) * self.DT
        new_dot = np.clip(new_dot, -self.MAX_SPEED, self.MAX_SPEED)
        self.theta = self.theta + new_dot * self.DT
        self.theta_dot = new_dot
        return (-cost).astype(np.float32), np.zeros(self.num_envs, dtype=bool), None
