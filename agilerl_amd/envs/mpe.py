"""Vectorized multi-agent particle environments (MPE-style).

First-party, batched reimplementations of the two benchmark scenarios the
reference is evaluated on (BASELINE config 4; reference uses PettingZoo
MPE): ``simple_speaker_listener`` and ``simple_spread``.  Point-mass
physics (velocity damping + acceleration actions) vectorized over all N
env instances; PettingZoo parallel-API-shaped dict observations/actions,
batched per agent.
"""

from __future__ import annotations

from typing import Dict, List, Optional

import numpy as np

from ..spaces import Box, Discrete, Space

__all__ = ["MultiAgentVecEnv", "SpeakerListenerVecEnv", "SimpleSpreadVecEnv"]

DT = 0.1
DAMPING = 0.25
ACCEL = 5.0
MAX_SPEED = 1.3


class MultiAgentVecEnv:
    """Base: PettingZoo-parallel-shaped API, batched over N instances."""

    agents: List[str]
    observation_spaces: Dict[str, Space]
    action_spaces: Dict[str, Space]
    max_episode_steps: int = 25

    def __init__(self, num_envs: int = 1, seed: Optional[int] = None):
        self.num_envs = int(num_envs)
        self.rng = np.random.default_rng(seed)
        self._elapsed = np.zeros(self.num_envs, dtype=np.int64)

    @property
    def possible_agents(self) -> List[str]:
        return self.agents

    def observation_space(self, agent: str) -> Space:
        return self.observation_spaces[agent]

    def action_space(self, agent: str) -> Space:
        return self.action_spaces[agent]

    # hooks ------------------------------------------------------------
    def _reset_rows(self, mask: np.ndarray) -> None:
        raise NotImplementedError

    def _obs(self) -> Dict[str, np.ndarray]:
        raise NotImplementedError

    def _step_all(self, actions: Dict[str, np.ndarray]) -> Dict[str, np.ndarray]:
        """Returns per-agent rewards dict."""
        raise NotImplementedError

    # API ---------------------------------------------------------------
    def reset(self, seed: Optional[int] = None):
        if seed is not None:
            self.rng = np.random.default_rng(seed)
        self._reset_rows(np.ones(self.num_envs, dtype=bool))
        self._elapsed[:] = 0
        return self._obs(), {}

    TERMINATES_AT_LIMIT = False  # probe envs: episode END is a true terminal

    def step(self, actions: Dict[str, np.ndarray]):
        rewards = self._step_all(actions)
        self._elapsed += 1
        limit = self._elapsed >= self.max_episode_steps
        if self.TERMINATES_AT_LIMIT:
            term_arr, trunc_arr = limit, np.zeros(self.num_envs, dtype=bool)
        else:
            term_arr, trunc_arr = np.zeros(self.num_envs, dtype=bool), limit
        term = {a: term_arr.copy() for a in self.agents}
        trunc = {a: trunc_arr.copy() for a in self.agents}
        obs = self._obs()
        info: Dict = {}
        done_arr = term_arr | trunc_arr
        if done_arr.any():
            info["final_observation"] = {a: o.copy() for a, o in obs.items()}
            self._reset_rows(done_arr)
            self._elapsed[done_arr] = 0
            obs = self._obs()
        return obs, rewards, term, trunc, info

    def close(self) -> None:
        pass

    @staticmethod
    def _move(vel: np.ndarray, u: np.ndarray) -> np.ndarray:
        """One physics step for a point mass: damped velocity + acceleration."""
        vel = vel * (1.0 - DAMPING) + u * ACCEL * DT
        speed = np.linalg.norm(vel, axis=-1, keepdims=True)
        scale = np.where(speed > MAX_SPEED, MAX_SPEED / np.maximum(speed, 1e-8), 1.0)
        return vel * scale

    @staticmethod
    def _discrete_to_force(a: np.ndarray) -> np.ndarray:
        """Discrete(5) -> 2D unit force: [noop, -x, +x, -y, +y]."""
        force = np.zeros((*a.shape, 2))
        force[..., 0] = np.where(a == 1, -1.0, np.where(a == 2, 1.0, 0.0))
        force[..., 1] = np.where(a == 3, -1.0, np.where(a == 4, 1.0, 0.0))
        return force


class SpeakerListenerVecEnv(MultiAgentVecEnv):
    """simple_speaker_listener: speaker sees the goal color and talks; the
    listener moves to the goal landmark it cannot see directly.
    Shared reward: -||listener - goal landmark||."""

    N_LANDMARKS = 3

    def __init__(self, num_envs: int = 1, seed: Optional[int] = None, continuous_actions: bool = False):
        super().__init__(num_envs, seed)
        self.continuous_actions = continuous_actions
        self.agents = ["speaker_0", "listener_0"]
        self.observation_spaces = {
            "speaker_0": Box(-np.inf, np.inf, (3,)),
            "listener_0": Box(-np.inf, np.inf, (11,)),
        }
        if continuous_actions:
            self.action_spaces = {
                "speaker_0": Box(0.0, 1.0, (3,)),
                "listener_0": Box(0.0, 1.0, (5,)),
            }
        else:
            self.action_spaces = {"speaker_0": Discrete(3), "listener_0": Discrete(5)}
        N = self.num_envs
        self.listener_pos = np.zeros((N, 2))
        self.listener_vel = np.zeros((N, 2))
        self.landmarks = np.zeros((N, self.N_LANDMARKS, 2))
        self.goal = np.zeros(N, dtype=np.int64)
        self.comm = np.zeros((N, 3))

    def _reset_rows(self, mask: np.ndarray) -> None:
        n = int(mask.sum())
        self.listener_pos[mask] = self.rng.uniform(-1, 1, (n, 2))
        self.listener_vel[mask] = 0.0
        self.landmarks[mask] = self.rng.uniform(-1, 1, (n, self.N_LANDMARKS, 2))
        self.goal[mask] = self.rng.integers(0, self.N_LANDMARKS, n)
        self.comm[mask] = 0.0

    def _obs(self) -> Dict[str, np.ndarray]:
        N = self.num_envs
        goal_onehot = np.zeros((N, 3))
        goal_onehot[np.arange(N), self.goal] = 1.0
        rel = self.landmarks - self.listener_pos[:, None, :]
        listener_obs = np.concatenate(
            [self.listener_vel, rel.reshape(N, -1), self.comm], axis=1
        )
        return {
            "speaker_0": goal_onehot.astype(np.float32),
            "listener_0": listener_obs.astype(np.float32),
        }

    def _step_all(self, actions):
        sp = np.asarray(actions["speaker_0"])
        li = np.asarray(actions["listener_0"])
        if self.continuous_actions:
            self.comm = sp.reshape(self.num_envs, 3).astype(np.float64)
            move_logits = li.reshape(self.num_envs, 5)
            u = np.stack(
                [move_logits[:, 2] - move_logits[:, 1], move_logits[:, 4] - move_logits[:, 3]],
                axis=1,
            )
        else:
            comm = np.zeros((self.num_envs, 3))
            comm[np.arange(self.num_envs), sp.reshape(-1).astype(np.int64)] = 1.0
            self.comm = comm
            u = self._discrete_to_force(li.reshape(-1).astype(np.int64))
        self.listener_vel = self._move(self.listener_vel, u)
        self.listener_pos = self.listener_pos + self.listener_vel * DT

        goal_pos = self.landmarks[np.arange(self.num_envs), self.goal]
        dist = np.linalg.norm(self.listener_pos - goal_pos, axis=1)
        reward = (-dist).astype(np.float32)
        return {a: reward.copy() for a in self.agents}


class SimpleSpreadVecEnv(MultiAgentVecEnv):
    """simple_spread: 3 agents cover 3 landmarks; global reward is the
    negative sum of (min agent distance to each landmark) minus collision
    penalties."""

    N_AGENTS = 3
    N_LANDMARKS = 3
    AGENT_SIZE = 0.15

    def __init__(self, num_envs: int = 1, seed: Optional[int] = None, continuous_actions: bool = False):
        super().__init__(num_envs, seed)
        self.continuous_actions = continuous_actions
        self.agents = [f"agent_{i}" for i in range(self.N_AGENTS)]
        obs_dim = 4 + 2 * self.N_LANDMARKS + 2 * (self.N_AGENTS - 1)
        self.observation_spaces = {a: Box(-np.inf, np.inf, (obs_dim,)) for a in self.agents}
        if continuous_actions:
            self.action_spaces = {a: Box(0.0, 1.0, (5,)) for a in self.agents}
        else:
            self.action_spaces = {a: Discrete(5) for a in self.agents}
        N = self.num_envs
        self.pos = np.zeros((N, self.N_AGENTS, 2))
        self.vel = np.zeros((N, self.N_AGENTS, 2))
        self.landmarks = np.zeros((N, self.N_LANDMARKS, 2))

    def _reset_rows(self, mask: np.ndarray) -> None:
        n = int(mask.sum())
        self.pos[mask] = self.rng.uniform(-1, 1, (n, self.N_AGENTS, 2))
        self.vel[mask] = 0.0
        self.landmarks[mask] = self.rng.uniform(-1, 1, (n, self.N_LANDMARKS, 2))

    def _obs(self) -> Dict[str, np.ndarray]:
        N = self.num_envs
        out = {}
        for i, name in enumerate(self.agents):
            rel_lm = (self.landmarks - self.pos[:, i : i + 1, :]).reshape(N, -1)
            others = [j for j in range(self.N_AGENTS) if j != i]
            rel_ag = (self.pos[:, others, :] - self.pos[:, i : i + 1, :]).reshape(N, -1)
            out[name] = np.concatenate(
                [self.vel[:, i], self.pos[:, i], rel_lm, rel_ag], axis=1
            ).astype(np.float32)
        return out

    def _step_all(self, actions):
        for i, name in enumerate(self.agents):
            a = np.asarray(actions[name])
            if self.continuous_actions:
                logits = a.reshape(self.num_envs, 5)
                u = np.stack(
                    [logits[:, 2] - logits[:, 1], logits[:, 4] - logits[:, 3]], axis=1
                )
            else:
                u = self._discrete_to_force(a.reshape(-1).astype(np.int64))
            self.vel[:, i] = self._move(self.vel[:, i], u)
            self.pos[:, i] = self.pos[:, i] + self.vel[:, i] * DT

        # global reward: -sum over landmarks of min agent distance
        d = np.linalg.norm(
            self.pos[:, :, None, :] - self.landmarks[:, None, :, :], axis=-1
        )  # (N, n_agents, n_landmarks)
        reward = -d.min(axis=1).sum(axis=1)
        # collision penalty
        pd = np.linalg.norm(self.pos[:, :, None, :] - self.pos[:, None, :, :], axis=-1)
        ii = np.arange(self.N_AGENTS)
        pd[:, ii, ii] = np.inf
        collisions = (pd < 2 * self.AGENT_SIZE).sum(axis=(1, 2)) / 2
        reward = (reward - collisions).astype(np.float32)
        return {a: reward.copy() for a in self.agents}
