"""Determinism: same seeds produce identical training trajectories."""

import numpy as np
import torch

from agilerl_amd.algorithms import DQN, PPO
from agilerl_amd.components import ReplayBuffer, RolloutBuffer
from agilerl_amd.envs import CartPoleVecEnv
from agilerl_amd.rollouts.on_policy import collect_rollouts

NET = {"arch": "mlp", "hidden_size": [16]}


def _dqn_run():
    np.random.seed(7), torch.manual_seed(7)
    env = CartPoleVecEnv(num_envs=4, seed=7)
    agent = DQN(env.observation_space, env.action_space, net_config=dict(NET),
                batch_size=32, lr=1e-3)
    buf = ReplayBuffer(500)
    obs, _ = env.reset()
    for _ in range(40):
        action = agent.get_action(obs, epsilon=0.3)
        next_obs, reward, term, trunc, _ = env.step(action)
        buf.add(obs=obs, action=action, reward=reward, next_obs=next_obs,
                done=term.astype(np.float32))
        obs = next_obs
    losses = [agent.learn(buf.sample(32)) for _ in range(10)]
    probe = torch.linspace(-1, 1, 16).reshape(4, 4)
    return losses, agent.actor(probe).detach().numpy()


def _ppo_run():
    np.random.seed(9), torch.manual_seed(9)
    env = CartPoleVecEnv(num_envs=4, seed=9)
    agent = PPO(env.observation_space, env.action_space, net_config=dict(NET),
                learn_step=16, batch_size=32)
    buf = RolloutBuffer(16, 4, gamma=agent.gamma, gae_lambda=agent.gae_lambda)
    collect_rollouts(agent, env, buf, 16)
    stats = agent.learn(buf)
    probe = torch.linspace(-1, 1, 16).reshape(4, 4)
    return stats, agent.actor(agent.actor.preprocess(probe)).detach().numpy()


def test_dqn_trajectory_is_deterministic():
    l1, q1 = _dqn_run()
    l2, q2 = _dqn_run()
    np.testing.assert_array_equal(np.array(l1), np.array(l2))
    np.testing.assert_array_equal(q1, q2)


def test_ppo_trajectory_is_deterministic():
    s1, p1 = _ppo_run()
    s2, p2 = _ppo_run()
    assert s1["policy_loss"] == s2["policy_loss"]
    np.testing.assert_array_equal(p1, p2)
