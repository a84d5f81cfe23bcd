"""PettingZoo-parallel env wrappers.

Reference parity: ``agilerl/wrappers/pettingzoo_wrappers.py:18``
(auto-reset parallel wrapper).  Wraps a single (non-vectorized)
PettingZoo-style parallel env so that an all-agents-done step
automatically resets — the contract the async vectorizer and the
multi-agent loops expect.
"""

from __future__ import annotations

from typing import Any, Dict, Optional

__all__ = ["AutoResetParallelWrapper"]


class AutoResetParallelWrapper:
    def __init__(self, env):
        self.env = env

    def __getattr__(self, name):
        return getattr(self.env, name)

    @property
    def possible_agents(self):
        return self.env.possible_agents

    def observation_space(self, agent):
        return self.env.observation_space(agent)

    def action_space(self, agent):
        return self.env.action_space(agent)

    def reset(self, seed: Optional[int] = None):
        return self.env.reset(seed=seed)

    def step(self, actions: Dict[str, Any]):
        obs, rewards, terms, truncs, info = self.env.step(actions)
        agents = self.env.possible_agents
        done = all(terms.get(a, False) or truncs.get(a, False) for a in agents)
        if done:
            info = dict(info or {})
            info["final_observation"] = obs
            obs, _ = self.env.reset()
        return obs, rewards, terms, truncs, info
