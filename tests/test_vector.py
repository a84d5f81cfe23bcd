"""Async/serial vector-env wrapper tests (subprocess + shared memory)."""

import numpy as np
import pytest

from agilerl_amd.spaces import Box, Discrete


class _ToySingleEnv:
    """Deterministic counter env for worker testing."""

    observation_space = Box(-np.inf, np.inf, (2,))
    action_space = Discrete(2)

    def __init__(self):
        self.t = 0

    def reset(self, seed=None):
        self.t = 0
        return np.array([self.t, 0.0], dtype=np.float32), {}

    def step(self, action):
        self.t += 1
        obs = np.array([self.t, float(action)], dtype=np.float32)
        term = self.t >= 5
        return obs, float(action), term, False, {}


class _BrokenEnv(_ToySingleEnv):
    def step(self, action):
        raise ValueError("boom")


class _ToyPZEnv:
    possible_agents = ["a0", "a1"]

    def __init__(self):
        self.t = 0
        self._spaces = {a: Box(-np.inf, np.inf, (2,)) for a in self.possible_agents}

    def observation_space(self, agent):
        return self._spaces[agent]

    def action_space(self, agent):
        return Discrete(2)

    def reset(self, seed=None):
        self.t = 0
        return {a: np.array([0.0, i], dtype=np.float32) for i, a in enumerate(self.possible_agents)}, {}

    def step(self, actions):
        self.t += 1
        obs = {a: np.array([self.t, actions[a]], dtype=np.float32) for a in self.possible_agents}
        rewards = {a: float(actions[a]) for a in self.possible_agents}
        terms = {a: self.t >= 3 for a in self.possible_agents}
        truncs = {a: False for a in self.possible_agents}
        return obs, rewards, terms, truncs, {}


class TestDummyVecEnv:
    def test_roundtrip(self):
        from agilerl_amd.vector import DummyVecEnv

        env = DummyVecEnv([_ToySingleEnv for _ in range(3)])
        obs, _ = env.reset(seed=0)
        assert obs.shape == (3, 2)
        for t in range(6):
            obs, r, te, tr, info = env.step(np.ones(3, dtype=int))
        assert te.any() or obs[:, 0].max() <= 5


class TestAsyncVectorEnv:
    def test_shared_memory_obs(self):
        from agilerl_amd.vector import AsyncVectorEnv

        env = AsyncVectorEnv([_ToySingleEnv for _ in range(4)])
        try:
            obs, _ = env.reset(seed=0)
            assert obs.shape == (4, 2)
            obs, r, te, tr, info = env.step(np.array([0, 1, 0, 1]))
            np.testing.assert_allclose(obs[:, 0], 1.0)
            np.testing.assert_allclose(r, [0, 1, 0, 1])
            # run to auto-reset
            for _ in range(5):
                obs, r, te, tr, info = env.step(np.zeros(4, dtype=int))
            assert "final_observation" in info or obs[:, 0].max() < 5
        finally:
            env.close()


class TestAsyncPZVecEnv:
    def test_dict_api(self):
        from agilerl_amd.vector import AsyncPettingZooVecEnv

        env = AsyncPettingZooVecEnv([_ToyPZEnv for _ in range(3)])
        try:
            obs, _ = env.reset()
            assert obs["a0"].shape == (3, 2)
            actions = {"a0": np.ones(3, dtype=int), "a1": np.zeros(3, dtype=int)}
            obs, rewards, terms, truncs, _ = env.step(actions)
            np.testing.assert_allclose(rewards["a0"], 1.0)
            np.testing.assert_allclose(rewards["a1"], 0.0)
            np.testing.assert_allclose(obs["a0"][:, 0], 1.0)
        finally:
            env.close()

    def test_worker_error_propagates(self):
        from agilerl_amd.vector import AsyncVectorEnv

        env = AsyncVectorEnv([_BrokenEnv for _ in range(2)])
        env.reset()
        with pytest.raises(RuntimeError, match="crashed"):
            env.step(np.zeros(2, dtype=int))
