"""Synthetic visual control env (Pong-like, image observations).

BASELINE config 3 names Rainbow-DQN on Atari Pong; ALE ROMs are not
available offline, so this env provides the same pipeline shape —
(4, 84, 84) uint8 frame-stack observations, small discrete action set,
sparse +-1 rewards — with vectorized numpy rendering: a ball falls with
a random horizontal drift and the paddle must catch it.
"""

from __future__ import annotations

from typing import Optional

import numpy as np

from ..spaces import Box, Discrete
from .base import BatchedVecEnv

__all__ = ["CatchPongVecEnv"]


class CatchPongVecEnv(BatchedVecEnv):
    max_episode_steps = 2000

    H = W = 84
    FRAMES = 4
    PADDLE_W = 12
    BALL = 3

    def __init__(self, num_envs: int = 1, seed: Optional[int] = None):
        super().__init__(num_envs, seed)
        self.single_observation_space = Box(
            0, 255, (self.FRAMES, self.H, self.W), dtype=np.uint8
        )
        self.single_action_space = Discrete(3)  # noop / left / right
        N = self.num_envs
        self.ball_x = np.zeros(N)
        self.ball_y = np.zeros(N)
        self.ball_vx = np.zeros(N)
        self.paddle_x = np.zeros(N)
        self.frames = np.zeros((N, self.FRAMES, self.H, self.W), dtype=np.uint8)

    def _spawn_ball(self, mask: np.ndarray) -> None:
        n = int(mask.sum())
        self.ball_x[mask] = self.rng.uniform(5, self.W - 5, n)
        self.ball_y[mask] = 2.0
        self.ball_vx[mask] = self.rng.uniform(-1.0, 1.0, n)

    def _reset_rows(self, mask: np.ndarray) -> None:
        self._spawn_ball(mask)
        self.paddle_x[mask] = self.W / 2
        self.frames[mask] = 0
        self._render(mask_all=True)

    def _render(self, mask_all: bool = False) -> None:
        """Shift the frame stack and draw the new frame."""
        self.frames[:, :-1] = self.frames[:, 1:]
        frame = np.zeros((self.num_envs, self.H, self.W), dtype=np.uint8)
        bx = self.ball_x.astype(int).clip(0, self.W - self.BALL)
        by = self.ball_y.astype(int).clip(0, self.H - self.BALL)
        px = self.paddle_x.astype(int).clip(0, self.W - self.PADDLE_W)
        for d in range(self.BALL):
            for e in range(self.BALL):
                frame[np.arange(self.num_envs), by + d, bx + e] = 255
        rows = np.arange(self.num_envs)[:, None]
        cols = px[:, None] + np.arange(self.PADDLE_W)[None, :]
        frame[rows, self.H - 3, cols] = 180
        frame[rows, self.H - 2, cols] = 180
        self.frames[:, -1] = frame

    def _obs(self) -> np.ndarray:
        return self.frames.copy()

    def _step_all(self, actions: np.ndarray):
        a = actions.reshape(-1)
        self.paddle_x += np.where(a == 1, -3.0, 0.0) + np.where(a == 2, 3.0, 0.0)
        self.paddle_x = self.paddle_x.clip(0, self.W - self.PADDLE_W)
        self.ball_y += 2.0
        self.ball_x += self.ball_vx
        bounce = (self.ball_x <= 0) | (self.ball_x >= self.W - self.BALL)
        self.ball_vx = np.where(bounce, -self.ball_vx, self.ball_vx)
        self.ball_x = self.ball_x.clip(0, self.W - self.BALL)

        at_bottom = self.ball_y >= self.H - 5
        caught = at_bottom & (
            (self.ball_x + self.BALL >= self.paddle_x)
            & (self.ball_x <= self.paddle_x + self.PADDLE_W)
        )
        missed = at_bottom & ~caught
        reward = np.where(caught, 1.0, np.where(missed, -1.0, 0.0)).astype(np.float32)
        # respawn ball after a catch or miss; episode ends after a miss
        if at_bottom.any():
            self._spawn_ball(at_bottom)
        self._render()
        return reward, missed, None
