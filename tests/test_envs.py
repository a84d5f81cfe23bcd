"""First-party batched env tests."""

import numpy as np
import pytest

from agilerl_amd.envs import CartPoleVecEnv, LunarLanderVecEnv, PendulumVecEnv, make_vect_envs


class TestCartPole:
    def test_api(self):
        env = CartPoleVecEnv(num_envs=4, seed=0)
        obs, info = env.reset()
        assert obs.shape == (4, 4)
        for _ in range(10):
            obs, r, term, trunc, info = env.step(np.random.randint(0, 2, 4))
            assert obs.shape == (4, 4)
            assert r.shape == (4,)
        assert env.single_action_space.n == 2

    def test_auto_reset_and_final_obs(self):
        env = CartPoleVecEnv(num_envs=8, seed=0)
        env.reset()
        saw_done = False
        for _ in range(300):
            obs, r, term, trunc, info = env.step(np.random.randint(0, 2, 8))
            done = term | trunc
            if done.any():
                saw_done = True
                assert "final_observation" in info
                # reset rows should be near the origin again
                assert np.all(np.abs(obs[done][:, 0]) < 0.06)
                break
        assert saw_done

    def test_determinism_with_seed(self):
        e1 = CartPoleVecEnv(4, seed=7)
        e2 = CartPoleVecEnv(4, seed=7)
        o1, _ = e1.reset()
        o2, _ = e2.reset()
        np.testing.assert_array_equal(o1, o2)


class TestLunarLander:
    def test_api_and_termination(self):
        env = LunarLanderVecEnv(num_envs=8, seed=0)
        obs, _ = env.reset()
        assert obs.shape == (8, 8)
        total_done = 0
        for _ in range(1200):
            obs, r, term, trunc, info = env.step(np.random.randint(0, 4, 8))
            total_done += int((term | trunc).sum())
            assert np.isfinite(r).all()
            assert np.isfinite(obs).all()
        assert total_done > 0  # random policy crashes eventually

    def test_landing_reward_sign(self):
        """Main-engine hovering straight down should beat free-fall crash."""
        env = LunarLanderVecEnv(num_envs=4, seed=0)
        env.reset()
        crash_rewards = []
        for _ in range(600):
            _, r, term, _, _ = env.step(np.zeros(4, dtype=int))  # noop -> crash
            crash_rewards.extend(r[term].tolist())
            if term.any():
                break
        assert crash_rewards and min(crash_rewards) < -50


class TestPendulum:
    def test_api(self):
        env = PendulumVecEnv(num_envs=4, seed=0)
        obs, _ = env.reset()
        assert obs.shape == (4, 3)
        obs, r, term, trunc, _ = env.step(np.random.uniform(-2, 2, (4, 1)))
        assert (r <= 0).all()


def test_registry():
    env = make_vect_envs("CartPole-v1", num_envs=2)
    assert env.num_envs == 2
    with pytest.raises(KeyError):
        make_vect_envs("NoSuchEnv-v0")
