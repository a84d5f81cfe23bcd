"""Multi-agent analytic probe environments.

Reference parity: ``agilerl/utils/probe_envs_ma.py`` (multi-agent
constant/obs-dependent/policy probes with known Q values, used to verify
centralized-critic learning).  Batched dict API like the MPE envs.
"""

from __future__ import annotations

from typing import Optional

import numpy as np

from ..spaces import Box, Discrete
from .mpe import MultiAgentVecEnv

__all__ = ["ConstantRewardMAEnv", "FixedObsPolicyMAEnv", "JointActionMAEnv"]


class _ProbeMA(MultiAgentVecEnv):
    max_episode_steps = 1
    TERMINATES_AT_LIMIT = True
    N_AGENTS = 2

    def __init__(self, num_envs: int = 1, seed: Optional[int] = None, continuous_actions: bool = False):
        super().__init__(num_envs, seed)
        self.agents = [f"agent_{i}" for i in range(self.N_AGENTS)]
        if continuous_actions:
            self.action_spaces = {a: Box(-1.0, 1.0, (1,)) for a in self.agents}
        else:
            self.action_spaces = {a: Discrete(2) for a in self.agents}
        self.observation_spaces = {a: Box(0.0, 1.0, (1,)) for a in self.agents}
        self.continuous_actions = continuous_actions

    def _reset_rows(self, mask):
        pass

    def _obs(self):
        return {a: np.zeros((self.num_envs, 1), dtype=np.float32) for a in self.agents}


class ConstantRewardMAEnv(_ProbeMA):
    """Both agents always receive +1.  Q*(s, a_joint) = 1."""

    q_values = 1.0

    def _step_all(self, actions):
        r = np.ones(self.num_envs, dtype=np.float32)
        return {a: r.copy() for a in self.agents}


class FixedObsPolicyMAEnv(_ProbeMA):
    """Per-agent reward: +1 for its own action 0, -1 otherwise (independent)."""

    def _step_all(self, actions):
        out = {}
        for a in self.agents:
            act = np.asarray(actions[a]).reshape(self.num_envs, -1)
            if self.continuous_actions:
                out[a] = (-((act[:, 0] - 0.5) ** 2)).astype(np.float32)
            else:
                out[a] = np.where(act[:, 0] == 0, 1.0, -1.0).astype(np.float32)
        return out


class JointActionMAEnv(_ProbeMA):
    """Shared reward +1 iff BOTH agents pick action 0 — needs the
    centralized critic to model the joint action."""

    def _step_all(self, actions):
        a0 = np.asarray(actions[self.agents[0]]).reshape(-1)
        a1 = np.asarray(actions[self.agents[1]]).reshape(-1)
        r = np.where((a0 == 0) & (a1 == 0), 1.0, -1.0).astype(np.float32)
        return {a: r.copy() for a in self.agents}
