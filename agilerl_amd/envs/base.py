"""First-party vectorized environment API.

The reference uses Gymnasium ``AsyncVectorEnv`` (one subprocess per env;
``agilerl/utils/utils.py:222``).  MI355X-native design: environments are
**natively batched** — one numpy/torch program steps all N env instances at
once (vectorized physics), which removes per-env subprocess+pipe overhead
entirely and keeps the obs batch in one contiguous array ready for pinned
host staging.  The API mirrors Gymnasium's vector API so the training
loops read familiarly:

    obs, info = env.reset(seed)
    obs, reward, terminated, truncated, info = env.step(actions)

Auto-reset: when an episode ends the returned ``obs`` row is the reset
observation and ``info["final_observation"]`` carries the true terminal
observation (for correct truncation bootstrapping).
"""

from __future__ import annotations

from typing import Any, Dict, Optional, Tuple

import numpy as np

from ..spaces import Space

__all__ = ["VecEnv", "BatchedVecEnv"]


class VecEnv:
    """Abstract vectorized env."""

    num_envs: int
    single_observation_space: Space
    single_action_space: Space

    @property
    def observation_space(self) -> Space:
        return self.single_observation_space

    @property
    def action_space(self) -> Space:
        return self.single_action_space

    def reset(self, seed: Optional[int] = None) -> Tuple[np.ndarray, Dict[str, Any]]:
        raise NotImplementedError

    def step(self, actions) -> Tuple[np.ndarray, np.ndarray, np.ndarray, np.ndarray, Dict[str, Any]]:
        raise NotImplementedError

    def close(self) -> None:
        pass

    def render(self):  # pragma: no cover
        return None


def _copy_obs(obs):
    """Deep-ish copy of a (possibly dict/tuple structured) observation."""
    if isinstance(obs, dict):
        return {k: _copy_obs(v) for k, v in obs.items()}
    if isinstance(obs, (tuple, list)):
        return type(obs)(_copy_obs(v) for v in obs)
    return obs.copy() if hasattr(obs, "copy") else obs


class BatchedVecEnv(VecEnv):
    """Base for natively-batched numpy envs with auto-reset.

    Subclasses implement ``_reset_rows(mask)`` (reset the masked env rows
    in internal state) and ``_step_all(actions)`` returning
    ``(obs, reward, terminated, truncated)`` arrays.
    """

    max_episode_steps: Optional[int] = None

    def __init__(self, num_envs: int, seed: Optional[int] = None):
        self.num_envs = int(num_envs)
        self.rng = np.random.default_rng(seed)
        self._elapsed = np.zeros(self.num_envs, dtype=np.int64)
        self._ep_return = np.zeros(self.num_envs, dtype=np.float64)

    # -- subclass hooks -------------------------------------------------
    def _reset_rows(self, mask: np.ndarray) -> None:
        raise NotImplementedError

    def _obs(self) -> np.ndarray:
        raise NotImplementedError

    def _step_all(self, actions: np.ndarray) -> Tuple[np.ndarray, np.ndarray, np.ndarray]:
        """Returns (reward, terminated, extra_info_dict_or_None)."""
        raise NotImplementedError

    # -- API ------------------------------------------------------------
    def reset(self, seed: Optional[int] = None):
        if seed is not None:
            self.rng = np.random.default_rng(seed)
        self._reset_rows(np.ones(self.num_envs, dtype=bool))
        self._elapsed[:] = 0
        self._ep_return[:] = 0.0
        return self._obs(), {}

    def step(self, actions):
        actions = np.asarray(actions)
        reward, terminated, info = self._step_all(actions)
        info = info or {}
        self._elapsed += 1
        truncated = (
            self._elapsed >= self.max_episode_steps
            if self.max_episode_steps is not None
            else np.zeros(self.num_envs, dtype=bool)
        )
        truncated = truncated & ~terminated
        done = terminated | truncated
        obs = self._obs()
        self._ep_return += reward
        if done.any():
            info = dict(info)
            info["final_observation"] = _copy_obs(obs)
            info["episode_return"] = self._ep_return[done].copy()
            self._reset_rows(done)
            self._elapsed[done] = 0
            self._ep_return[done] = 0.0
            obs = self._obs()
        return obs, reward, terminated, truncated, info
