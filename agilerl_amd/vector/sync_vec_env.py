"""Serial vectorizer over per-env factories.

Reference parity: gym.vector.SyncVectorEnv as used by
``agilerl/utils/utils.py:222`` (``make_vect_envs(should_async_vector=
False)``).  Same 5-tuple contract as :class:`AsyncVectorEnv` — batched
obs/reward/term/trunc plus ``info["final_observation"]`` on autoreset —
without subprocess workers (right for cheap Python envs where IPC would
dominate).
"""

from __future__ import annotations

from typing import Any, Callable, Dict, List, Optional

import numpy as np

__all__ = ["SyncVectorEnv"]


class SyncVectorEnv:
    def __init__(self, env_fns: List[Callable], copy: bool = True):
        self.copy = bool(copy)
        self.envs = [fn() for fn in env_fns]
        self.num_envs = len(self.envs)
        self.single_observation_space = self.envs[0].observation_space
        self.single_action_space = self.envs[0].action_space

    def reset(self, seed: Optional[int] = None):
        obs = []
        for i, env in enumerate(self.envs):
            kw = {} if seed is None else {"seed": seed + i}
            o, _ = env.reset(**kw)
            obs.append(np.asarray(o))
        return np.stack(obs).astype(np.float32, copy=self.copy), {}

    def step(self, actions):
        actions = np.asarray(actions)
        obs, rewards, terms, truncs = [], [], [], []
        finals: List[Optional[np.ndarray]] = []
        for env, a in zip(self.envs, actions):
            o, r, te, tr, _ = env.step(a.item() if np.ndim(a) == 0 else a)
            if te or tr:
                finals.append(np.asarray(o))
                o, _ = env.reset()
            else:
                finals.append(None)
            obs.append(np.asarray(o))
            rewards.append(r)
            terms.append(te)
            truncs.append(tr)
        obs_arr = np.stack(obs).astype(np.float32, copy=self.copy)
        info: Dict[str, Any] = {}
        if any(f is not None for f in finals):
            fo = obs_arr.copy()
            for i, f in enumerate(finals):
                if f is not None:
                    fo[i] = f
            info["final_observation"] = fo
        return (obs_arr, np.asarray(rewards, dtype=np.float32),
                np.asarray(terms), np.asarray(truncs), info)

    def close(self):
        for env in self.envs:
            if hasattr(env, "close"):
                env.close()
