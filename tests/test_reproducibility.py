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


def _rainbow_run():
    from agilerl_amd.algorithms import RainbowDQN
    from agilerl_amd.components import PrioritizedReplayBuffer

    np.random.seed(3), torch.manual_seed(3)
    env = CartPoleVecEnv(num_envs=4, seed=3)
    agent = RainbowDQN(env.observation_space, env.action_space,
                       net_config=dict(NET), batch_size=32, lr=1e-3)
    buf = PrioritizedReplayBuffer(500)
    obs, _ = env.reset()
    for _ in range(30):
        action = agent.get_action(obs)
        next_obs, reward, term, trunc, _ = env.step(action)
        buf.add(obs=obs, action=action, reward=reward, next_obs=next_obs,
                done=term.astype(np.float32))
        obs = next_obs
    losses = []
    for _ in range(6):
        batch = buf.sample(32, beta=0.4)
        loss = agent.learn(batch)
        buf.update_priorities(batch["idxs"], agent.last_td_errors)
        losses.append(loss)
    return losses


def test_rainbow_per_trajectory_is_deterministic():
    l1 = _rainbow_run()
    l2 = _rainbow_run()
    np.testing.assert_array_equal(np.array(l1), np.array(l2))


def test_grpo_generation_is_deterministic():
    from agilerl_amd.algorithms.llm.grpo import GRPO
    from agilerl_amd.llm_envs import TokenReasoningGym

    tiny = dict(model_type="llama", vocab_size=64, hidden_size=32,
                intermediate_size=64, num_hidden_layers=1, num_attention_heads=2,
                num_key_value_heads=1, max_position_embeddings=64, pad_token_id=0)

    def run():
        np.random.seed(5), torch.manual_seed(5)
        agent = GRPO(model_config=dict(tiny), dtype=torch.float32,
                     lora_config={"r": 2}, group_size=2, max_completion_tokens=5)
        env = TokenReasoningGym(vocab_size=64, prompt_len=4, data_batch_size=2,
                                group_size=2, seed=5)
        return agent.get_action(env.reset())

    s1, s2 = run(), run()
    torch.testing.assert_close(s1, s2)
