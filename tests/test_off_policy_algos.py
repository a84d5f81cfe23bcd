"""Rainbow / DDPG / TD3 / CQN tests (probe convergence + mechanics)."""

import numpy as np
import torch

from agilerl_amd.algorithms import CQN, DDPG, TD3, RainbowDQN
from agilerl_amd.components import MultiStepReplayBuffer, PrioritizedReplayBuffer, ReplayBuffer
from agilerl_amd.envs.probe import (
    ConstantRewardContActionsEnv,
    ConstantRewardEnv,
    FixedObsPolicyContActionsEnv,
)
from agilerl_amd.spaces import Box, Discrete
from tests.test_algorithms import fill_buffer


class TestRainbow:
    def test_learn_with_per(self):
        env = ConstantRewardEnv(num_envs=4)
        agent = RainbowDQN(env.observation_space, env.action_space, lr=1e-3,
                           batch_size=32, v_min=-2, v_max=2,
                           net_config={"arch": "mlp", "hidden_size": [32]})
        buf = PrioritizedReplayBuffer(1000, alpha=0.6)
        fill_buffer(env, buf, 100)
        for i in range(100):
            batch = buf.sample(32, beta=0.5)
            loss = agent.learn(batch)
            buf.update_priorities(batch["idxs"], agent.last_td_errors)
        assert np.isfinite(loss)
        # Q(s) should approach 1 for both actions
        q = agent.actor(torch.zeros(1, 1))
        assert torch.allclose(q, torch.ones(1, 2), atol=0.35)

    def test_nstep_sampling(self):
        env = ConstantRewardEnv(num_envs=4)
        agent = RainbowDQN(env.observation_space, env.action_space, n_step=3,
                           batch_size=16, net_config={"arch": "mlp", "hidden_size": [16]})
        buf = MultiStepReplayBuffer(1000, n_step=3, gamma=agent.gamma)
        fill_buffer(env, buf, 60)
        batch = buf.sample(16)
        loss = agent.learn(batch)
        assert np.isfinite(loss)

    def test_noisy_exploration_no_epsilon(self):
        env = ConstantRewardEnv(num_envs=4)
        agent = RainbowDQN(env.observation_space, env.action_space)
        obs = np.zeros((4, 1), dtype=np.float32)
        a = agent.get_action(obs, training=True)
        assert a.shape == (4,)

    def test_clone_checkpoint(self, tmp_path):
        agent = RainbowDQN(Box(-1, 1, (4,)), Discrete(3))
        clone = agent.clone(index=2)
        x = torch.randn(2, 4)
        agent.actor.eval(), clone.actor.eval()
        assert torch.allclose(agent.actor(x), clone.actor(x))
        p = str(tmp_path / "r.pt")
        agent.save_checkpoint(p)
        loaded = RainbowDQN.load(p)
        loaded.actor.eval()
        assert torch.allclose(agent.actor(x), loaded.actor(x))


class TestDDPG:
    def test_probe_constant_reward(self):
        env = ConstantRewardContActionsEnv(num_envs=4)
        agent = DDPG(env.observation_space, env.action_space, lr_actor=1e-3,
                     lr_critic=1e-2, tau=0.1, batch_size=64,
                     net_config={"arch": "mlp", "hidden_size": [32]})
        buf = ReplayBuffer(2000)
        fill_buffer(env, buf, 200)
        for _ in range(300):
            agent.learn(buf.sample(64))
        obs = torch.zeros(1, 1)
        act = torch.zeros(1, 1)
        q = agent.critic(obs, act)
        assert torch.allclose(q, torch.ones(1, 1), atol=0.15)

    def test_probe_policy(self):
        env = FixedObsPolicyContActionsEnv(num_envs=4)
        agent = DDPG(env.observation_space, env.action_space, lr_actor=1e-3,
                     lr_critic=1e-2, tau=0.1, batch_size=64,
                     net_config={"arch": "mlp", "hidden_size": [32]})
        buf = ReplayBuffer(2000)
        fill_buffer(env, buf, 300)
        for _ in range(600):
            agent.learn(buf.sample(64))
        action = agent.get_action(np.zeros((1, 1), dtype=np.float32), training=False)
        assert abs(action[0, 0] - 0.5) < 0.15

    def test_ou_noise_reset(self):
        agent = DDPG(Box(-1, 1, (3,)), Box(-1, 1, (2,)))
        agent.get_action(np.zeros((4, 3), dtype=np.float32))
        assert agent._ou_state.shape == (4, 2)
        agent.reset_action_noise([1, 2])
        assert np.all(agent._ou_state[1] == 0)


class TestTD3:
    def test_learn_and_delayed_policy(self):
        env = ConstantRewardContActionsEnv(num_envs=4)
        agent = TD3(env.observation_space, env.action_space, batch_size=32,
                    policy_freq=2, net_config={"arch": "mlp", "hidden_size": [16]})
        buf = ReplayBuffer(500)
        fill_buffer(env, buf, 60)
        w_before = [p.detach().clone() for p in agent.actor.parameters()]
        agent.learn(buf.sample(32))  # counter=1: no policy update
        unchanged = all(
            torch.equal(b, p.detach()) for b, p in zip(w_before, agent.actor.parameters())
        )
        assert unchanged
        agent.learn(buf.sample(32))  # counter=2: policy update fires
        changed = any(
            not torch.equal(b, p.detach()) for b, p in zip(w_before, agent.actor.parameters())
        )
        assert changed

    def test_clone_has_twin(self):
        agent = TD3(Box(-1, 1, (4,)), Box(-1, 1, (2,)))
        clone = agent.clone(index=1)
        x, a = torch.randn(2, 4), torch.randn(2, 2)
        assert torch.allclose(agent.critic_2(x, a), clone.critic_2(x, a))

    def test_mutation_consistency(self):
        agent = TD3(Box(-1, 1, (4,)), Box(-1, 1, (2,)))
        agent.apply_architecture_mutation("encoder.add_node", numb_new_nodes=16)
        x, a = torch.randn(2, 4), torch.randn(2, 2)
        assert torch.allclose(agent.critic_2(x, a), agent.critic_2_target(x, a))
        assert torch.allclose(agent.actor(x), agent.actor_target(x))


class TestCQN:
    def test_offline_learn(self):
        env = ConstantRewardEnv(num_envs=4)
        agent = CQN(env.observation_space, env.action_space, batch_size=32,
                    net_config={"arch": "mlp", "hidden_size": [16]})
        buf = ReplayBuffer(500)
        fill_buffer(env, buf, 60)
        for _ in range(20):
            loss = agent.learn(buf.sample(32))
        assert np.isfinite(loss)
        assert agent.algo == "CQN"


def test_rainbow_cnn_on_visual_env():
    """Rainbow + PER + CNN encoder through a real uint8 visual env
    (BreakoutLite) — the Atari-like BASELINE config shape on CPU."""
    from agilerl_amd.algorithms import RainbowDQN
    from agilerl_amd.components import PrioritizedReplayBuffer
    from agilerl_amd.envs.visual import BreakoutLiteVecEnv

    torch.manual_seed(0), np.random.seed(0)
    env = BreakoutLiteVecEnv(num_envs=2, seed=0)
    agent = RainbowDQN(
        env.observation_space, env.action_space,
        net_config={"arch": "cnn", "channel_size": [8, 8],
                    "kernel_size": [8, 4], "stride_size": [4, 2]},
        batch_size=8, n_step=3,
    )
    buf = PrioritizedReplayBuffer(200, n_step=3)
    obs, _ = env.reset()
    for _ in range(12):
        action = agent.get_action(obs)
        next_obs, r, term, trunc, _ = env.step(action)
        buf.add(obs=obs, action=action, reward=r, next_obs=next_obs,
                done=term.astype(np.float32))
        obs = next_obs
    for _ in range(2):
        batch = buf.sample(8, beta=0.4)
        loss = agent.learn(batch)
        buf.update_priorities(batch["idxs"], agent.last_td_errors)
        assert np.isfinite(loss)
