"""Vectorized LunarLander (Box2D-free rigid-body reimplementation).

Same interface as the classic LunarLander task: 8-dim observation
``[x, y, vx, vy, angle, v_angle, leg1, leg2]`` (identically scaled),
4 discrete actions (noop / left engine / main engine / right engine), and
the same potential-based shaping reward (distance + speed + tilt terms,
leg-contact bonuses, fuel costs, ±100 terminal).  Dynamics are a direct
planar rigid-body integration (gravity, thrust along the body axis, side
thruster torque) rather than a Box2D simulation, vectorized over all N
instances in numpy — no per-env subprocesses.
"""

from __future__ import annotations

from typing import Optional

import numpy as np

from ..spaces import Box, Discrete
from .base import BatchedVecEnv

__all__ = ["LunarLanderVecEnv", "LunarLanderContinuousVecEnv"]


class LunarLanderVecEnv(BatchedVecEnv):
    max_episode_steps = 1000

    DT = 0.02
    GRAVITY = 10.0
    MAIN_ACCEL = 15.0
    SIDE_ACCEL = 1.5
    SIDE_TORQUE = 3.0
    ANGLE_DAMP = 0.99
    X_WORLD = 10.0  # half width (m)
    Y_WORLD = 20.0 / 3.0  # obs scale for y
    PAD_HALF_W = 1.5
    LEG_Y = 0.1  # contact height

    def __init__(self, num_envs: int = 1, seed: Optional[int] = None):
        super().__init__(num_envs, seed)
        high = np.full(8, np.inf, dtype=np.float32)
        self.single_observation_space = Box(-high, high)
        self.single_action_space = Discrete(4)
        # state: px, py, vx, vy, angle, vangle  (meters / rad)
        self.state = np.zeros((self.num_envs, 6), dtype=np.float64)
        self.legs = np.zeros((self.num_envs, 2), dtype=np.float64)
        self.prev_shaping = np.zeros(self.num_envs, dtype=np.float64)

    # ------------------------------------------------------------------
    def _reset_rows(self, mask: np.ndarray) -> None:
        n = int(mask.sum())
        s = np.zeros((n, 6))
        s[:, 0] = self.rng.uniform(-0.5, 0.5, n)  # px
        s[:, 1] = 10.0  # py
        s[:, 2] = self.rng.uniform(-2.0, 2.0, n)  # vx
        s[:, 3] = self.rng.uniform(-1.5, 0.5, n)  # vy
        s[:, 4] = self.rng.uniform(-0.15, 0.15, n)  # angle
        s[:, 5] = self.rng.uniform(-0.3, 0.3, n)  # vangle
        self.state[mask] = s
        self.legs[mask] = 0.0
        self.prev_shaping[mask] = self._shaping()[mask]

    def _obs(self) -> np.ndarray:
        px, py, vx, vy, ang, vang = self.state.T
        obs = np.stack(
            [
                px / self.X_WORLD,
                py / self.Y_WORLD,
                vx / 5.0,
                vy / 7.5,
                ang,
                0.4 * vang,
                self.legs[:, 0],
                self.legs[:, 1],
            ],
            axis=1,
        )
        return obs.astype(np.float32)

    def _shaping(self) -> np.ndarray:
        o = self._obs().astype(np.float64)
        return (
            -100.0 * np.sqrt(o[:, 0] ** 2 + o[:, 1] ** 2)
            - 100.0 * np.sqrt(o[:, 2] ** 2 + o[:, 3] ** 2)
            - 100.0 * np.abs(o[:, 4])
            + 10.0 * o[:, 6]
            + 10.0 * o[:, 7]
        )

    def _step_all(self, actions: np.ndarray):
        a = actions.reshape(-1).astype(np.int64)
        px, py, vx, vy, ang, vang = self.state.T

        main = (a == 2).astype(np.float64)
        left = (a == 1).astype(np.float64)  # fires left engine -> push right + spin
        right = (a == 3).astype(np.float64)

        # thrust along body up-axis (angle measured from vertical)
        ax = -np.sin(ang) * self.MAIN_ACCEL * main + (right - left) * self.SIDE_ACCEL * np.cos(ang)
        ay = np.cos(ang) * self.MAIN_ACCEL * main - self.GRAVITY + (right - left) * self.SIDE_ACCEL * np.sin(ang)
        aang = (left - right) * self.SIDE_TORQUE

        vx = vx + ax * self.DT
        vy = vy + ay * self.DT
        vang = (vang + aang * self.DT) * self.ANGLE_DAMP
        px = px + vx * self.DT
        py = py + vy * self.DT
        ang = ang + vang * self.DT

        # ground interaction — crash/landing judged on the RAW impact state;
        # shaping is also computed pre-clamp so zeroing the velocity at ground
        # contact cannot leak a spurious positive speed-term delta
        on_ground = py <= self.LEG_Y
        upright = np.abs(ang) < 0.4
        self.legs[:, 0] = (on_ground & upright).astype(np.float64)
        self.legs[:, 1] = (on_ground & upright).astype(np.float64)
        py = np.maximum(py, 0.0)
        grounded = py <= 0.0 + 1e-9
        impact_speed = np.sqrt(vx**2 + vy**2)

        self.state = np.stack([px, py, vx, vy, ang, vang], axis=1)
        shaping = self._shaping()
        reward = (shaping - self.prev_shaping).astype(np.float32)
        self.prev_shaping = shaping
        reward -= (0.30 * main + 0.03 * (left + right)).astype(np.float32)

        # now damp ground-contact velocities for rows that keep running
        vy = np.where(grounded & (vy < 0), 0.0, vy)
        vx = np.where(grounded, vx * 0.8, vx)
        vang = np.where(grounded, vang * 0.5, vang)
        self.state = np.stack([px, py, vx, vy, ang, vang], axis=1)

        crash = (grounded & (~upright | (impact_speed > 1.5))) | (np.abs(px) > self.X_WORLD)
        landed = grounded & upright & (impact_speed <= 1.5) & (np.abs(vang) < 0.3)
        terminated = crash | landed
        reward = np.where(crash, reward - 100.0, reward)
        reward = np.where(landed, reward + 100.0, reward)
        return reward.astype(np.float32), terminated, None


class LunarLanderContinuousVecEnv(LunarLanderVecEnv):
    """Continuous-action variant (LunarLanderContinuous-v2/-v3 interface):
    action (2,) in [-1, 1] — [main throttle, lateral].  Standard mapping:
    the main engine fires for a0 > 0 with power 0.5 + 0.5*a0; the lateral
    thruster fires when |a1| > 0.5 (sign picks the side), matching the
    classic env's control contract on the same first-party dynamics."""

    def __init__(self, num_envs: int = 1, seed: Optional[int] = None):
        super().__init__(num_envs, seed)
        self.single_action_space = Box(-1.0, 1.0, (2,))

    def _step_all(self, actions: np.ndarray):
        a = np.asarray(actions, dtype=np.float64).reshape(self.num_envs, 2)
        a = np.clip(a, -1.0, 1.0)
        px, py, vx, vy, ang, vang = self.state.T

        main = np.where(a[:, 0] > 0.0, 0.5 + 0.5 * a[:, 0], 0.0)
        lateral = np.where(np.abs(a[:, 1]) > 0.5, a[:, 1], 0.0)
        right = np.clip(lateral, 0.0, 1.0)
        left = np.clip(-lateral, 0.0, 1.0)

        ax = -np.sin(ang) * self.MAIN_ACCEL * main + (right - left) * self.SIDE_ACCEL * np.cos(ang)
        ay = np.cos(ang) * self.MAIN_ACCEL * main - self.GRAVITY + (right - left) * self.SIDE_ACCEL * np.sin(ang)
        aang = (left - right) * self.SIDE_TORQUE

        vx = vx + ax * self.DT
        vy = vy + ay * self.DT
        vang = (vang + aang * self.DT) * self.ANGLE_DAMP
        px = px + vx * self.DT
        py = py + vy * self.DT
        ang = ang + vang * self.DT

        on_ground = py <= self.LEG_Y
        upright = np.abs(ang) < 0.4
        self.legs[:, 0] = (on_ground & upright).astype(np.float64)
        self.legs[:, 1] = (on_ground & upright).astype(np.float64)
        py = np.maximum(py, 0.0)
        grounded = py <= 0.0 + 1e-9
        impact_speed = np.sqrt(vx**2 + vy**2)

        self.state = np.stack([px, py, vx, vy, ang, vang], axis=1)
        shaping = self._shaping()
        reward = (shaping - self.prev_shaping).astype(np.float32)
        self.prev_shaping = shaping
        reward -= (0.30 * main + 0.03 * np.abs(lateral)).astype(np.float32)

        vy = np.where(grounded & (vy < 0), 0.0, vy)
        vx = np.where(grounded, vx * 0.8, vx)
        vang = np.where(grounded, vang * 0.5, vang)
        self.state = np.stack([px, py, vx, vy, ang, vang], axis=1)

        crash = (grounded & (~upright | (impact_speed > 1.5))) | (np.abs(px) > self.X_WORLD)
        landed = grounded & upright & (impact_speed <= 1.5) & (np.abs(vang) < 0.3)
        terminated = crash | landed
        reward = np.where(crash, reward - 100.0, reward)
        reward = np.where(landed, reward + 100.0, reward)
        return reward.astype(np.float32), terminated, None
