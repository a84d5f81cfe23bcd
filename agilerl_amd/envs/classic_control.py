"""Vectorized MountainCar and Acrobot (batched numpy dynamics).

Completes the classic-control set (CartPole/Pendulum live in their own
modules).  Physics follow the standard published formulations for these
tasks; everything is computed rowwise over the batch with auto-reset
handled by BatchedVecEnv.
"""

from __future__ import annotations

from typing import Optional

import numpy as np

from ..spaces import Box, Discrete
from .base import BatchedVecEnv

__all__ = ["MountainCarVecEnv", "MountainCarContinuousVecEnv", "AcrobotVecEnv"]


class MountainCarVecEnv(BatchedVecEnv):
    """Discrete mountain car: push left / none / right; reward -1 per step
    until the car reaches the right hilltop (position >= 0.5)."""

    max_episode_steps = 200

    MIN_POS, MAX_POS = -1.2, 0.6
    MAX_SPEED = 0.07
    GOAL_POS = 0.5
    FORCE = 0.001
    GRAVITY = 0.0025

    def __init__(self, num_envs: int = 1, seed: Optional[int] = None):
        super().__init__(num_envs, seed)
        self.single_observation_space = Box(
            np.array([self.MIN_POS, -self.MAX_SPEED], dtype=np.float32),
            np.array([self.MAX_POS, self.MAX_SPEED], dtype=np.float32),
            (2,),
        )
        self.single_action_space = Discrete(3)
        self.pos = np.zeros(self.num_envs)
        self.vel = np.zeros(self.num_envs)

    def _reset_rows(self, mask):
        n = int(mask.sum())
        self.pos[mask] = self.rng.uniform(-0.6, -0.4, n)
        self.vel[mask] = 0.0

    def _obs(self):
        return np.stack([self.pos, self.vel], axis=1).astype(np.float32)

    def _force(self, actions) -> np.ndarray:
        return (np.asarray(actions).reshape(-1) - 1) * self.FORCE

    def _step_all(self, actions):
        self.vel += self._force(actions) + np.cos(3 * self.pos) * (-self.GRAVITY)
        self.vel = np.clip(self.vel, -self.MAX_SPEED, self.MAX_SPEED)
        self.pos = np.clip(self.pos + self.vel, self.MIN_POS, self.MAX_POS)
        self.vel[(self.pos <= self.MIN_POS) & (self.vel < 0)] = 0.0
        terminated = self.pos >= self.GOAL_POS
        reward = np.full(self.num_envs, -1.0, dtype=np.float32)
        return reward, terminated, None


class MountainCarContinuousVecEnv(MountainCarVecEnv):
    """Continuous-force variant: reward +100 on goal minus action cost."""

    max_episode_steps = 999
    POWER = 0.0015

    def __init__(self, num_envs: int = 1, seed: Optional[int] = None):
        super().__init__(num_envs, seed)
        self.single_action_space = Box(-1.0, 1.0, (1,))
        self.GOAL_POS = 0.45
        self._last_action = np.zeros(self.num_envs)

    def _force(self, actions) -> np.ndarray:
        a = np.clip(np.asarray(actions, dtype=np.float64).reshape(self.num_envs, -1)[:, 0], -1, 1)
        self._last_action = a
        return a * self.POWER

    def _step_all(self, actions):
        _, terminated, _ = super()._step_all(actions)
        reward = np.where(terminated, 100.0, 0.0) - 0.1 * self._last_action**2
        return reward.astype(np.float32), terminated, None


class AcrobotVecEnv(BatchedVecEnv):
    """Two-link underactuated pendulum; torque on the second joint; reward
    -1 per step until the tip swings above the bar
    (-cos(th1) - cos(th1 + th2) > 1)."""

    max_episode_steps = 500

    DT = 0.2
    L1 = L2 = 1.0
    M1 = M2 = 1.0
    LC1 = LC2 = 0.5
    I1 = I2 = 1.0
    G = 9.8
    MAX_VEL1 = 4 * np.pi
    MAX_VEL2 = 9 * np.pi
    TORQUES = np.array([-1.0, 0.0, 1.0])

    def __init__(self, num_envs: int = 1, seed: Optional[int] = None):
        super().__init__(num_envs, seed)
        high = np.array([1, 1, 1, 1, self.MAX_VEL1, self.MAX_VEL2], dtype=np.float32)
        self.single_observation_space = Box(-high, high, (6,))
        self.single_action_space = Discrete(3)
        self.state = np.zeros((self.num_envs, 4))  # th1, th2, dth1, dth2

    def _reset_rows(self, mask):
        n = int(mask.sum())
        self.state[mask] = self.rng.uniform(-0.1, 0.1, (n, 4))

    def _obs(self):
        th1, th2, d1, d2 = self.state.T
        return np.stack(
            [np.cos(th1), np.sin(th1), np.cos(th2), np.sin(th2), d1, d2], axis=1
        ).astype(np.float32)

    def _dsdt(self, s: np.ndarray, torque: np.ndarray) -> np.ndarray:
        th1, th2, dth1, dth2 = s.T
        m1, m2, l1, lc1, lc2, i1, i2, g = (
            self.M1, self.M2, self.L1, self.LC1, self.LC2, self.I1, self.I2, self.G,
        )
        d1 = m1 * lc1**2 + m2 * (l1**2 + lc2**2 + 2 * l1 * lc2 * np.cos(th2)) + i1 + i2
        d2 = m2 * (lc2**2 + l1 * lc2 * np.cos(th2)) + i2
        phi2 = m2 * lc2 * g * np.cos(th1 + th2 - np.pi / 2)
        phi1 = (
            -m2 * l1 * lc2 * dth2**2 * np.sin(th2)
            - 2 * m2 * l1 * lc2 * dth2 * dth1 * np.sin(th2)
            + (m1 * lc1 + m2 * l1) * g * np.cos(th1 - np.pi / 2)
            + phi2
        )
        ddth2 = (
            torque + d2 / d1 * phi1 - m2 * l1 * lc2 * dth1**2 * np.sin(th2) - phi2
        ) / (m2 * lc2**2 + i2 - d2**2 / d1)
        ddth1 = -(d2 * ddth2 + phi1) / d1
        return np.stack([dth1, dth2, ddth1, ddth2], axis=1)

    def _step_all(self, actions):
        torque = self.TORQUES[np.asarray(actions).reshape(-1)]
        # rk4 over one DT step
        s = self.state
        k1 = self._dsdt(s, torque)
        k2 = self._dsdt(s + 0.5 * self.DT * k1, torque)
        k3 = self._dsdt(s + 0.5 * self.DT * k2, torque)
        k4 = self._dsdt(s + self.DT * k3, torque)
        s = s + self.DT / 6.0 * (k1 + 2 * k2 + 2 * k3 + k4)
        s[:, 0] = (s[:, 0] + np.pi) % (2 * np.pi) - np.pi
        s[:, 1] = (s[:, 1] + np.pi) % (2 * np.pi) - np.pi
        s[:, 2] = np.clip(s[:, 2], -self.MAX_VEL1, self.MAX_VEL1)
        s[:, 3] = np.clip(s[:, 3], -self.MAX_VEL2, self.MAX_VEL2)
        self.state = s
        terminated = -np.cos(s[:, 0]) - np.cos(s[:, 0] + s[:, 1]) > 1.0
        reward = np.where(terminated, 0.0, -1.0).astype(np.float32)
        return reward, terminated, None
