"""Cooperative pong, batched multi-agent (PettingZoo butterfly
cooperative_pong_v6 stand-in).

Two paddles keep one ball in play; both agents receive the shared
survival reward and the episode ends when the ball leaves either side.
First-party vector-state dynamics (the offline image has no PettingZoo/
pygame): observations are low-dimensional states rather than frames,
with the same agent set, cooperative reward structure and Discrete(3)
actions (stay / up / down) as the reference env.
"""

from __future__ import annotations

from typing import Dict, List, Optional

import numpy as np

from ..spaces import Box, Discrete
from .mpe import MultiAgentVecEnv

__all__ = ["CooperativePongVecEnv"]

PADDLE_SPEED = 0.1
PADDLE_HALF = 0.15
BALL_SPEED = 0.06


class CooperativePongVecEnv(MultiAgentVecEnv):
    agents = ["paddle_0", "paddle_1"]
    max_episode_steps = 200
    TERMINATES_AT_LIMIT = False

    def __init__(self, num_envs: int = 1, seed: Optional[int] = None, **kwargs):
        super().__init__(num_envs, seed)
        obs_space = Box(-1.5, 1.5, (6,))
        self.observation_spaces = {a: obs_space for a in self.agents}
        self.action_spaces = {a: Discrete(3) for a in self.agents}
        N = self.num_envs
        self.ball = np.zeros((N, 2))
        self.vel = np.zeros((N, 2))
        self.pad_y = np.zeros((N, 2))  # left, right
        self._dead = np.zeros(N, dtype=bool)

    def _reset_rows(self, mask: np.ndarray) -> None:
        n = int(mask.sum())
        if n == 0:
            return
        self.ball[mask] = 0.0
        angle = self.rng.uniform(-0.7, 0.7, size=n)
        direction = np.where(self.rng.random(n) < 0.5, -1.0, 1.0)
        self.vel[mask, 0] = direction * BALL_SPEED * np.cos(angle)
        self.vel[mask, 1] = BALL_SPEED * np.sin(angle)
        self.pad_y[mask] = 0.0
        self._dead[mask] = False

    def _obs(self) -> Dict[str, np.ndarray]:
        base = np.concatenate(
            [self.ball, self.vel / BALL_SPEED, self.pad_y], axis=1
        ).astype(np.float32)
        # each paddle sees its own y first (index 4), partner's at 5
        flipped = base.copy()
        flipped[:, [4, 5]] = flipped[:, [5, 4]]
        return {"paddle_0": base, "paddle_1": flipped}

    def _step_all(self, actions: Dict[str, np.ndarray]) -> Dict[str, np.ndarray]:
        for i, aid in enumerate(self.agents):
            a = np.asarray(actions[aid]).reshape(-1)
            dy = np.where(a == 1, PADDLE_SPEED, np.where(a == 2, -PADDLE_SPEED, 0.0))
            self.pad_y[:, i] = np.clip(self.pad_y[:, i] + dy, -1.0, 1.0)

        self.ball += self.vel
        # top/bottom bounce
        hit_wall = np.abs(self.ball[:, 1]) > 1.0
        self.vel[hit_wall, 1] *= -1.0
        self.ball[:, 1] = np.clip(self.ball[:, 1], -1.0, 1.0)
        # paddle bounce at x = -1 (paddle 0) and x = +1 (paddle 1)
        for i, x_edge in ((0, -1.0), (1, 1.0)):
            at_edge = (self.ball[:, 0] * np.sign(x_edge)) >= 1.0
            saved = at_edge & (
                np.abs(self.ball[:, 1] - self.pad_y[:, i]) <= PADDLE_HALF
            )
            self.vel[saved, 0] *= -1.0
            # english: deflect by contact offset
            self.vel[saved, 1] += 0.3 * (self.ball[saved, 1] - self.pad_y[saved, i]) * BALL_SPEED
            self.ball[saved, 0] = x_edge * 0.999
            self._dead |= at_edge & ~saved
        reward = np.where(self._dead, 0.0, 0.1).astype(np.float32)
        return {a: reward.copy() for a in self.agents}

    def step(self, actions):
        obs, rewards, term, trunc, info = super().step(actions)
        # ball lost = true terminal for both agents
        if self._dead.any():
            dead = self._dead.copy()
            for a in self.agents:
                term[a] = term[a] | dead
            info.setdefault("final_observation", {a: o.copy() for a, o in obs.items()})
            self._reset_rows(dead)
            self._elapsed[dead] = 0
            obs = self._obs()
        return obs, rewards, term, trunc, info
