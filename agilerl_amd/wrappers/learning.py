"""Learning wrappers: curriculum skills and dataset->bandit adaptation.

Reference parity: ``agilerl/wrappers/learning.py`` — Skill :14 (curriculum
reward shaping wrapper), BanditEnv :66 (re-exported from envs.bandit).
"""

from __future__ import annotations

from ..envs.bandit import BanditEnv  # noqa: F401  (reference places it here)

__all__ = ["Skill", "BanditEnv"]


class Skill:
    """Curriculum skill: wraps a vectorized env and reshapes rewards /
    terminations to teach a sub-behavior.  Subclass and override
    ``skill_reward``; chain skills by training sequentially on each."""

    def __init__(self, env):
        self.env = env

    def __getattr__(self, name):
        return getattr(self.env, name)

    def skill_reward(self, obs, reward, terminated, truncated, info):
        """Override: return (reward, terminated, truncated) for the skill."""
        return reward, terminated, truncated

    def reset(self, *args, **kwargs):
        return self.env.reset(*args, **kwargs)

    def step(self, action):
        obs, reward, terminated, truncated, info = self.env.step(action)
        reward, terminated, truncated = self.skill_reward(
            obs, reward, terminated, truncated, info
        )
        return obs, reward, terminated, truncated, info
