"""Vectorized CartPole (classic-control dynamics, batched in numpy).

Same observation/action spaces and physics constants as the classic
CartPole-v1 task (pole-on-cart balancing; Barto, Sutton & Anderson 1983).
All N instances integrate in one vectorized step.
"""

from __future__ import annotations

from typing import Optional

import numpy as np

from ..spaces import Box, Discrete
from .base import BatchedVecEnv

__all__ = ["CartPoleVecEnv"]


class CartPoleVecEnv(BatchedVecEnv):
    max_episode_steps = 500

    GRAVITY = 9.8
    MASS_CART = 1.0
    MASS_POLE = 0.1
    TOTAL_MASS = MASS_CART + MASS_POLE
    LENGTH = 0.5  # half pole length
    POLEMASS_LENGTH = MASS_POLE * LENGTH
    FORCE_MAG = 10.0
    TAU = 0.02
    THETA_LIMIT = 12 * 2 * np.pi / 360
    X_LIMIT = 2.4

    def __init__(self, num_envs: int = 1, seed: Optional[int] = None):
        super().__init__(num_envs, seed)
        high = np.array(
            [self.X_LIMIT * 2, np.finfo(np.float32).max, self.THETA_LIMIT * 2, np.finfo(np.float32).max],
            dtype=np.float32,
        )
        self.single_observation_space = Box(-high, high)
        self.single_action_space = Discrete(2)
        self.state = np.zeros((self.num_envs, 4), dtype=np.float64)

    def _reset_rows(self, mask: np.ndarray) -> None:
        n = int(mask.sum())
        self.state[mask] = self.rng.uniform(-0.05, 0.05, size=(n, 4))

    def _obs(self) -> np.ndarray:
        return self.state.astype(np.float32)

    def _step_all(self, actions: np.ndarray):
        x, x_dot, theta, theta_dot = self.state.T
        force = np.where(actions.reshape(-1) == 1, self.FORCE_MAG, -self.FORCE_MAG)
        costheta = np.cos(theta)
        sintheta = np.sin(theta)
        temp = (force + self.POLEMASS_LENGTH * theta_dot**2 * sintheta) / self.TOTAL_MASS
        thetaacc = (self.GRAVITY * sintheta - costheta * temp) / (
            self.LENGTH * (4.0 / 3.0 - self.MASS_POLE * costheta**2 / self.TOTAL_MASS)
        )
        xacc = temp - self.POLEMASS_LENGTH * thetaacc * costheta / self.TOTAL_MASS
        x = x + self.TAU * x_dot
        x_dot = x_dot + self.TAU * xacc
        theta = theta + self.TAU * theta_dot
        theta_dot = theta_dot + self.TAU * thetaacc
        self.state = np.stack([x, x_dot, theta, theta_dot], axis=1)
        terminated = (
            (x < -self.X_LIMIT)
            | (x > self.X_LIMIT)
            | (theta < -self.THETA_LIMIT)
            | (theta > self.THETA_LIMIT)
        )
        reward = np.ones(self.num_envs, dtype=np.float32)
        return reward, terminated, None
