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

__all__ = ["CatchPongVecEnv", "BreakoutLiteVecEnv"]


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


class BreakoutLiteVecEnv(BatchedVecEnv):
    """Breakout-shaped visual env: paddle, ball, 4x8 brick wall; +1 per
    brick, episode ends on ball drop or wall cleared.  Same (4, 84, 84)
    uint8 frame-stack contract as CatchPong — a second Atari-like task so
    Rainbow/visual-PPO coverage is not single-game."""

    max_episode_steps = 2000

    H = W = 84
    FRAMES = 4
    PADDLE_W = 14
    BALL = 3
    ROWS, COLS = 4, 8
    BRICK_H = 4

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
        self.ball_vy = np.zeros(N)
        self.paddle_x = np.zeros(N)
        self.bricks = np.ones((N, self.ROWS, self.COLS), dtype=bool)
        self.frames = np.zeros((N, self.FRAMES, self.H, self.W), dtype=np.uint8)

    def _reset_rows(self, mask: np.ndarray) -> None:
        n = int(mask.sum())
        self.ball_x[mask] = self.rng.uniform(10, self.W - 10, n)
        self.ball_y[mask] = self.H / 2
        self.ball_vx[mask] = self.rng.choice([-1.2, 1.2], n)
        self.ball_vy[mask] = -1.5
        self.paddle_x[mask] = self.W / 2
        self.bricks[mask] = True
        self.frames[mask] = 0
        self._render()

    def _obs(self) -> np.ndarray:
        return self.frames.copy()

    def _brick_cell(self):
        bw = self.W // self.COLS
        col = (self.ball_x // bw).astype(int).clip(0, self.COLS - 1)
        row = ((self.ball_y - 6) // self.BRICK_H).astype(int)
        return row, col

    def _render(self) -> None:
        self.frames[:, :-1] = self.frames[:, 1:]
        frame = np.zeros((self.num_envs, self.H, self.W), dtype=np.uint8)
        bw = self.W // self.COLS
        for r in range(self.ROWS):
            y0 = 6 + r * self.BRICK_H
            for c in range(self.COLS):
                live = self.bricks[:, r, c]
                frame[live, y0 : y0 + self.BRICK_H - 1, c * bw : (c + 1) * bw - 1] = 128
        bx = self.ball_x.astype(int).clip(0, self.W - self.BALL)
        by = self.ball_y.astype(int).clip(0, self.H - self.BALL)
        for i in range(self.num_envs):
            frame[i, by[i] : by[i] + self.BALL, bx[i] : bx[i] + self.BALL] = 255
            p = int(np.clip(self.paddle_x[i] - self.PADDLE_W // 2, 0, self.W - self.PADDLE_W))
            frame[i, self.H - 3 : self.H - 1, p : p + self.PADDLE_W] = 255
        self.frames[:, -1] = frame

    def _step_all(self, actions):
        a = np.asarray(actions).reshape(-1)
        self.paddle_x += np.where(a == 1, -2.5, 0.0) + np.where(a == 2, 2.5, 0.0)
        self.paddle_x = np.clip(self.paddle_x, self.PADDLE_W / 2, self.W - self.PADDLE_W / 2)
        self.ball_x += self.ball_vx
        self.ball_y += self.ball_vy
        # walls
        hit_side = (self.ball_x <= 0) | (self.ball_x >= self.W - self.BALL)
        self.ball_vx = np.where(hit_side, -self.ball_vx, self.ball_vx)
        self.ball_x = np.clip(self.ball_x, 0, self.W - self.BALL)
        hit_top = self.ball_y <= 0
        self.ball_vy = np.where(hit_top, -self.ball_vy, self.ball_vy)
        self.ball_y = np.clip(self.ball_y, 0, None)
        # bricks
        reward = np.zeros(self.num_envs, dtype=np.float32)
        row, col = self._brick_cell()
        in_wall = (row >= 0) & (row < self.ROWS) & (self.ball_vy < 0)
        for i in np.flatnonzero(in_wall):
            if self.bricks[i, row[i], col[i]]:
                self.bricks[i, row[i], col[i]] = False
                self.ball_vy[i] = -self.ball_vy[i]
                reward[i] = 1.0
        # paddle bounce
        at_paddle = self.ball_y >= self.H - 5
        caught = at_paddle & (np.abs(self.ball_x + self.BALL / 2 - self.paddle_x) <= self.PADDLE_W / 2 + 1)
        self.ball_vy = np.where(caught & (self.ball_vy > 0), -self.ball_vy, self.ball_vy)
        dropped = (self.ball_y >= self.H - 2) & ~caught
        cleared = ~self.bricks.any(axis=(1, 2))
        terminated = dropped | cleared
        self._render()
        return reward, terminated, None
