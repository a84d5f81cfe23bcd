"""Buffer / segment-tree / ops numerics tests (CPU references)."""

import numpy as np
import pytest
import torch

from agilerl_amd import ops
from agilerl_amd.components import (
    MinSegmentTree,
    MultiStepReplayBuffer,
    PrioritizedReplayBuffer,
    ReplayBuffer,
    RolloutBuffer,
    SumSegmentTree,
)


class TestSegmentTree:
    def test_sum_and_retrieve(self):
        tree = SumSegmentTree(16)
        vals = torch.tensor([1.0, 2.0, 3.0, 4.0])
        tree.update(torch.arange(4), vals)
        assert tree.sum() == pytest.approx(10.0)
        # prefix 0 -> leaf 0; 0.9 -> leaf 0; 1.0 -> leaf 1; 5.9 -> leaf 2; 6.0 -> leaf 3
        idx = tree.retrieve(torch.tensor([0.0, 0.9, 1.0, 5.9, 6.0, 9.9]))
        assert idx.tolist() == [0, 0, 1, 2, 3, 3]

    def test_min_tree(self):
        tree = MinSegmentTree(8)
        tree.update(torch.arange(5), torch.tensor([5.0, 3.0, 8.0, 1.0, 9.0]))
        assert tree.min() == pytest.approx(1.0)
        tree.update(torch.tensor([3]), torch.tensor([7.0]))
        assert tree.min() == pytest.approx(3.0)

    def test_random_consistency(self):
        rng = np.random.default_rng(0)
        tree = SumSegmentTree(64)
        ref = np.zeros(64)
        for _ in range(10):
            idx = rng.choice(64, size=8, replace=False)
            vals = rng.random(8).astype(np.float32)
            tree.update(torch.from_numpy(idx), torch.from_numpy(vals))
            ref[idx] = vals
            assert tree.sum() == pytest.approx(ref.sum(), rel=1e-5)


class TestReplayBuffer:
    def test_add_sample(self):
        buf = ReplayBuffer(100)
        for t in range(20):
            buf.add(
                obs=np.random.rand(4, 3).astype(np.float32),
                action=np.random.randint(0, 2, 4),
                reward=np.random.rand(4).astype(np.float32),
                next_obs=np.random.rand(4, 3).astype(np.float32),
                done=np.zeros(4, dtype=np.float32),
            )
        assert len(buf) == 80
        batch = buf.sample(16)
        assert batch["obs"].shape == (16, 3)
        assert batch["action"].shape == (16,)

    def test_circular_overwrite(self):
        buf = ReplayBuffer(8)
        for t in range(5):
            buf.add(obs=np.full((4, 2), t, dtype=np.float32), action=np.zeros(4),
                    reward=np.zeros(4, dtype=np.float32), next_obs=np.zeros((4, 2), dtype=np.float32),
                    done=np.zeros(4, dtype=np.float32))
        assert len(buf) == 8
        # capacity 8 / 4-env adds: storage holds the newest two adds (t=3, t=4)
        vals = set(buf._storage["obs"][:, 0].tolist())
        assert vals == {3.0, 4.0}

    def test_dict_obs(self):
        buf = ReplayBuffer(50)
        buf.add(
            obs={"a": np.random.rand(4, 3), "b": np.random.rand(4, 2)},
            action=np.zeros(4),
            reward=np.zeros(4, dtype=np.float32),
            next_obs={"a": np.random.rand(4, 3), "b": np.random.rand(4, 2)},
            done=np.zeros(4, dtype=np.float32),
        )
        batch = buf.sample(2)
        assert batch["obs"]["a"].shape == (2, 3)


class TestNStep:
    def test_nstep_scan_reference(self):
        rewards = torch.tensor([[1.0, 1.0, 1.0], [1.0, 2.0, 4.0], [1.0, 1.0, 1.0]])
        dones = torch.tensor([[0.0, 0.0, 0.0], [0.0, 0.0, 0.0], [0.0, 1.0, 0.0]])
        gamma = 0.9
        returns, steps = ops.nstep_scan(rewards, dones, gamma)
        assert returns[0] == pytest.approx(1 + 0.9 + 0.81)
        assert returns[1] == pytest.approx(1 + 0.9 * 2 + 0.81 * 4)
        # done at k=1 cuts the window after step 1 (reward at k=1 included)
        assert returns[2] == pytest.approx(1 + 0.9 * 1)
        assert steps.tolist() == [3.0, 3.0, 2.0]

    def test_nstep_buffer(self):
        buf = MultiStepReplayBuffer(1000, n_step=3, gamma=0.9)
        num_envs = 2
        for t in range(50):
            buf.add(
                obs=np.full((num_envs, 1), t, dtype=np.float32),
                action=np.zeros(num_envs),
                reward=np.ones(num_envs, dtype=np.float32),
                next_obs=np.full((num_envs, 1), t + 1, dtype=np.float32),
                done=np.zeros(num_envs, dtype=np.float32),
            )
        batch = buf.sample(32)
        # no dones -> all 3-step windows with return 1+0.9+0.81
        assert torch.allclose(batch["reward"], torch.full((32,), 2.71), atol=1e-5)
        assert (batch["n_steps"] == 3).all()
        # next_obs should be obs + 3
        assert torch.allclose(batch["next_obs"][:, 0], batch["obs"][:, 0] + 3)


class TestPER:
    def test_priority_sampling_bias(self):
        buf = PrioritizedReplayBuffer(64, alpha=1.0)
        for t in range(32):
            buf.add(obs=np.array([[t]], dtype=np.float32), action=np.zeros(1),
                    reward=np.zeros(1, dtype=np.float32), next_obs=np.array([[t]], dtype=np.float32),
                    done=np.zeros(1, dtype=np.float32))
        # give index 7 a huge priority
        buf.update_priorities(torch.tensor([7]), torch.tensor([1000.0]))
        batch = buf.sample(256, beta=0.4)
        frac = (batch["obs"][:, 0] == 7).float().mean()
        assert frac > 0.8

    def test_weights_bounded(self):
        buf = PrioritizedReplayBuffer(64, alpha=0.6)
        for t in range(32):
            buf.add(obs=np.random.rand(1, 2), action=np.zeros(1),
                    reward=np.zeros(1, dtype=np.float32), next_obs=np.random.rand(1, 2),
                    done=np.zeros(1, dtype=np.float32))
        batch = buf.sample(16, beta=1.0)
        assert (batch["weights"] <= 1.0 + 1e-5).all()
        assert (batch["weights"] > 0).all()


class TestGAE:
    def test_matches_manual_scan(self):
        T, N = 12, 3
        torch.manual_seed(0)
        rewards = torch.randn(T, N)
        values = torch.randn(T, N)
        dones = (torch.rand(T, N) < 0.2).float()
        last_value = torch.randn(N)
        gamma, lam = 0.99, 0.95
        adv, ret = ops.gae_scan(rewards, values, dones, last_value, gamma, lam)
        # manual per-env python scan: dones[t] (done-after-step-t) cuts both
        # step t's bootstrap and its lambda carry
        for n in range(N):
            next_adv, next_val = 0.0, last_value[n].item()
            expected = np.zeros(T)
            for t in reversed(range(T)):
                nd = 1.0 - dones[t, n].item()
                delta = rewards[t, n].item() + gamma * next_val * nd - values[t, n].item()
                next_adv = delta + gamma * lam * nd * next_adv
                expected[t] = next_adv
                next_val = values[t, n].item()
            np.testing.assert_allclose(adv[:, n].numpy(), expected, rtol=1e-4, atol=1e-5)

    def test_terminal_step_advantage_is_r_minus_v(self):
        """A mid-rollout terminal step must not bootstrap the next episode's
        value: adv[t_term] == r[t_term] - V(s_t_term) exactly."""
        T, N = 8, 2
        torch.manual_seed(1)
        rewards = torch.randn(T, N)
        values = torch.randn(T, N)
        dones = torch.zeros(T, N)
        dones[3, 0] = 1.0
        dones[5, 1] = 1.0
        last_value = torch.randn(N)
        adv, _ = ops.gae_scan(rewards, values, dones, last_value, 0.99, 0.95)
        assert adv[3, 0].item() == pytest.approx(
            (rewards[3, 0] - values[3, 0]).item(), abs=1e-6
        )
        assert adv[5, 1].item() == pytest.approx(
            (rewards[5, 1] - values[5, 1]).item(), abs=1e-6
        )
        # and the step BEFORE the terminal still carries through it normally
        delta2 = rewards[2, 0] + 0.99 * values[3, 0] - values[2, 0]
        expect2 = delta2 + 0.99 * 0.95 * adv[3, 0]
        assert adv[2, 0].item() == pytest.approx(expect2.item(), abs=1e-5)

    def test_rollout_buffer_end_to_end(self):
        buf = RolloutBuffer(capacity=8, num_envs=4, gamma=0.99, gae_lambda=0.95)
        for t in range(8):
            buf.add(
                obs=np.random.rand(4, 3),
                action=np.random.randint(0, 2, 4),
                reward=np.random.rand(4),
                done=np.zeros(4),
                value=np.random.rand(4),
                log_prob=np.random.rand(4),
            )
        buf.compute_returns_and_advantages(torch.zeros(4))
        mbs = list(buf.get_minibatches(16))
        assert sum(mb["obs"].shape[0] for mb in mbs) == 32
        assert "advantages" in mbs[0]


class TestC51:
    def test_projection_conserves_mass(self):
        B, A = 8, 51
        dist = torch.softmax(torch.randn(B, A), dim=-1)
        support = torch.linspace(-10, 10, A)
        proj = ops.c51_project(dist, torch.randn(B), (torch.rand(B) < 0.5).float(), support, 0.99, -10, 10)
        assert torch.allclose(proj.sum(-1), torch.ones(B), atol=1e-5)

    def test_terminal_collapses_to_reward_atom(self):
        A = 51
        support = torch.linspace(-10, 10, A)
        dist = torch.full((1, A), 1.0 / A)
        proj = ops.c51_project(dist, torch.tensor([0.0]), torch.tensor([1.0]), support, 0.99, -10, 10)
        # done=1: target = reward = 0 -> all mass at atom 25 (z=0)
        assert proj[0, 25] == pytest.approx(1.0, abs=1e-5)


class TestPolyak:
    def test_lerp(self):
        a = [torch.zeros(3), torch.zeros(2)]
        b = [torch.ones(3), torch.ones(2)]
        ops.polyak_update_(a, b, 0.25)
        assert torch.allclose(a[0], torch.full((3,), 0.25))


class TestGroupAdvantage:
    def test_centering(self):
        r = torch.tensor([1.0, 2.0, 3.0, 10.0, 20.0, 30.0])
        adv = ops.group_advantage(r, group_size=3, scale=False)
        assert adv[:3].sum() == pytest.approx(0.0, abs=1e-5)
        assert adv[3:].sum() == pytest.approx(0.0, abs=1e-5)


class TestReferenceBufferSurface:
    """Reference buffer constructor kwargs (components/replay_buffer.py
    dtype, rollout_buffer.py use_gae/wrap_at_capacity)."""

    def test_replay_buffer_storage_dtype(self):
        import numpy as np
        import torch

        from agilerl_amd.components import ReplayBuffer

        b = ReplayBuffer(100, dtype=torch.float16)
        b.add(obs=np.random.randn(4, 3).astype(np.float32),
              action=np.zeros(4, np.int64), reward=np.ones(4, np.float32),
              next_obs=np.random.randn(4, 3).astype(np.float32),
              done=np.zeros(4, np.float32))
        s = b.sample(2)
        assert s["obs"].dtype == torch.float16
        assert s["action"].dtype == torch.int64  # ints untouched

    def test_rollout_use_gae_false_discounted_returns(self):
        import numpy as np
        import torch

        from agilerl_amd.components.rollout_buffer import RolloutBuffer

        buf = RolloutBuffer(3, num_envs=1, gamma=0.5, use_gae=False)
        for r in (1.0, 1.0, 1.0):
            buf.add(obs=np.zeros((1, 2), np.float32), action=np.zeros((1,), np.int64),
                    reward=np.array([r], np.float32), done=np.zeros(1, np.float32),
                    value=np.array([0.25], np.float32), log_prob=np.zeros(1, np.float32))
        buf.compute_returns_and_advantages(np.zeros(1, np.float32))
        # rewards-to-go with gamma 0.5: [1.75, 1.5, 1.0]
        assert torch.allclose(buf.returns.reshape(-1),
                              torch.tensor([1.75, 1.5, 1.0]))
        assert torch.allclose(buf.advantages, buf.returns - 0.25)

    def test_rollout_wrap_at_capacity(self):
        import numpy as np
        import pytest

        from agilerl_amd.components.rollout_buffer import RolloutBuffer

        def one(buf, r):
            buf.add(obs=np.zeros((1, 2), np.float32), action=np.zeros((1,), np.int64),
                    reward=np.array([r], np.float32), done=np.zeros(1, np.float32),
                    value=np.zeros(1, np.float32), log_prob=np.zeros(1, np.float32))

        strict = RolloutBuffer(2, num_envs=1)
        one(strict, 1.0); one(strict, 2.0)
        with pytest.raises(RuntimeError, match="wrap_at_capacity"):
            one(strict, 3.0)
        ring = RolloutBuffer(2, num_envs=1, wrap_at_capacity=True)
        one(ring, 1.0); one(ring, 2.0); one(ring, 3.0)  # overwrites slot 0
        assert float(ring._storage["reward"][0]) == 3.0

    def test_rsnorm_dict_norm_obs_keys(self):
        import numpy as np

        from agilerl_amd.spaces import Box, DictSpace
        from agilerl_amd.wrappers.agent import RSNorm

        class FakeAgent:
            observation_space = DictSpace({"v": Box(-1, 1, (3,)), "img": Box(0, 255, (2, 2))})

            def get_action(self, obs, training=True):
                return obs

        w = RSNorm(FakeAgent(), norm_obs_keys=["v"])
        obs = {"v": np.random.randn(4, 3).astype(np.float32) * 5 + 2,
               "img": np.zeros((4, 2, 2), np.float32)}
        for _ in range(50):
            w.get_action(obs)
        out = w.get_action(obs, training=False)
        assert abs(out["v"].mean()) < 1.0      # normalized
        assert (out["img"] == 0).all()         # untouched key
        st = w.wrapper_state()
        w2 = RSNorm(FakeAgent(), norm_obs_keys=["v"])
        w2.load_wrapper_state(st)
        assert np.allclose(w2.rms["v"].mean, w.rms["v"].mean)
