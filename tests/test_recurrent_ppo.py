"""Recurrent PPO (LSTM + BPTT) tests."""

import numpy as np
import torch

from agilerl_amd.algorithms.ppo import PPO
from agilerl_amd.components import RolloutBuffer
from agilerl_amd.envs import CartPoleVecEnv
from agilerl_amd.rollouts.on_policy import collect_rollouts_recurrent


def make_agent(**kw):
    env = CartPoleVecEnv(8, seed=0)
    agent = PPO(env.observation_space, env.action_space, recurrent=True,
                learn_step=32, batch_size=128, **kw)
    return env, agent


class TestRecurrentPPO:
    def test_lstm_encoder_selected(self):
        _, agent = make_agent()
        assert agent.actor.is_recurrent
        assert agent.critic.is_recurrent

    def test_hidden_threading(self):
        env, agent = make_agent()
        obs, _ = env.reset()
        hidden = agent.init_hidden(8)
        a, lp, v, hidden2 = agent.get_action_recurrent(obs, hidden)
        assert a.shape == (8,)
        assert hidden2["ha"].shape == hidden["ha"].shape
        assert not torch.equal(hidden2["ha"], hidden["ha"])

    def test_collect_and_bptt_learn(self):
        env, agent = make_agent(lr=1e-3)
        buf = RolloutBuffer(32, 8, gamma=agent.gamma, gae_lambda=agent.gae_lambda)
        obs = done = hidden = None
        obs, done, hidden, stats = collect_rollouts_recurrent(agent, env, buf, 32, obs, done, hidden)
        st = agent.learn(buf)
        assert np.isfinite(st["policy_loss"])
        assert np.isfinite(st["approx_kl"])

    def test_hidden_reset_on_done(self):
        env, agent = make_agent()
        obs, _ = env.reset()
        hidden = agent.init_hidden(8)
        # run until at least one episode ends
        for _ in range(300):
            a, lp, v, hidden = agent.get_action_recurrent(obs, hidden)
            obs, r, te, tr, _ = env.step(a)
            done = te | tr
            if done.any():
                mask = torch.as_tensor(~done, dtype=torch.float32)
                hidden = {k: h * mask.view(1, -1, 1) for k, h in hidden.items()}
                assert (hidden["ha"][:, done, :] == 0).all()
                break

    def test_mutation_and_clone(self):
        env, agent = make_agent()
        agent.apply_architecture_mutation("encoder.add_node", numb_new_nodes=16)
        assert agent.actor.encoder.hidden_state_size == agent.critic.encoder.hidden_state_size
        clone = agent.clone(index=1)
        obs, _ = env.reset()
        h = clone.init_hidden(8)
        a, lp, v, _ = clone.get_action_recurrent(obs, h)
        assert a.shape == (8,)

    def test_learning_progress_smoke(self):
        torch.manual_seed(0)
        np.random.seed(0)
        env, agent = make_agent(lr=2e-3, ent_coef=0.0)
        buf = RolloutBuffer(32, 8, gamma=agent.gamma, gae_lambda=agent.gae_lambda)
        obs = done = hidden = None
        for _ in range(10):
            obs, done, hidden, stats = collect_rollouts_recurrent(
                agent, env, buf, 32, obs, done, hidden
            )
            agent.learn(buf)
        fit = agent.test(env, loop=1)
        assert np.isfinite(fit)
