"""Analytic probe environments for learning-correctness tests.

Reference parity: ``agilerl/utils/probe_envs.py`` (ConstantRewardEnv :29 …
PolicyContActionsEnv :966).  Each env has known correct Q / V / policy
values; tests train briefly and assert convergence.
All are vectorized (BatchedVecEnv) with episode length 1 or 2.
"""

from __future__ import annotations

from typing import Optional

import numpy as np

from ..spaces import Box, Discrete
from .base import BatchedVecEnv

__all__ = [
    "ConstantRewardEnv",
    "ObsDependentRewardEnv",
    "DiscountedRewardEnv",
    "FixedObsPolicyEnv",
    "PolicyEnv",
    "ConstantRewardContActionsEnv",
    "FixedObsPolicyContActionsEnv",
]


class ConstantRewardEnv(BatchedVecEnv):
    """Always reward 1, one step.  Q*(s, a) = 1 for all a."""

    max_episode_steps = 1
    q_values = np.array([[1.0, 1.0]])
    v_values = np.array([[1.0]])

    def __init__(self, num_envs: int = 1, seed: Optional[int] = None):
        super().__init__(num_envs, seed)
        self.single_observation_space = Box(0.0, 1.0, (1,))
        self.single_action_space = Discrete(2)

    def _reset_rows(self, mask):
        pass

    def _obs(self):
        return np.zeros((self.num_envs, 1), dtype=np.float32)

    def _step_all(self, actions):
        return (
            np.ones(self.num_envs, dtype=np.float32),
            np.ones(self.num_envs, dtype=bool),
            None,
        )


class ObsDependentRewardEnv(BatchedVecEnv):
    """Obs 0 -> reward -1, obs 1 -> reward +1, one step.

    Q*(s=0, a)=-1, Q*(s=1, a)=+1.
    """

    max_episode_steps = 1
    q_values = np.array([[-1.0, -1.0], [1.0, 1.0]])
    v_values = np.array([[-1.0], [1.0]])

    def __init__(self, num_envs: int = 1, seed: Optional[int] = None):
        super().__init__(num_envs, seed)
        self.single_observation_space = Box(0.0, 1.0, (1,))
        self.single_action_space = Discrete(2)
        self.state = np.zeros(self.num_envs, dtype=np.float32)

    def _reset_rows(self, mask):
        self.state[mask] = self.rng.integers(0, 2, int(mask.sum())).astype(np.float32)

    def _obs(self):
        return self.state.reshape(-1, 1).copy()

    def _step_all(self, actions):
        reward = np.where(self.state > 0.5, 1.0, -1.0).astype(np.float32)
        return reward, np.ones(self.num_envs, dtype=bool), None


class DiscountedRewardEnv(BatchedVecEnv):
    """Two steps: obs 0 then obs 1; reward 1 only on the second step.

    Q*(s=0) = gamma, Q*(s=1) = 1 — tests the discount/bootstrap path.
    """

    max_episode_steps = 2

    def __init__(self, num_envs: int = 1, seed: Optional[int] = None):
        super().__init__(num_envs, seed)
        self.single_observation_space = Box(0.0, 1.0, (1,))
        self.single_action_space = Discrete(2)
        self.phase = np.zeros(self.num_envs, dtype=np.float32)

    def q_values_for(self, gamma: float) -> np.ndarray:
        return np.array([[gamma, gamma], [1.0, 1.0]])

    def _reset_rows(self, mask):
        self.phase[mask] = 0.0

    def _obs(self):
        return self.phase.reshape(-1, 1).copy()

    def _step_all(self, actions):
        second = self.phase > 0.5
        reward = np.where(second, 1.0, 0.0).astype(np.float32)
        terminated = second.copy()
        self.phase = np.where(second, self.phase, 1.0)
        return reward, terminated, None


class FixedObsPolicyEnv(BatchedVecEnv):
    """Fixed obs; action 0 -> +1, action 1 -> -1.  Optimal policy: action 0."""

    max_episode_steps = 1
    q_values = np.array([[1.0, -1.0]])

    def __init__(self, num_envs: int = 1, seed: Optional[int] = None):
        super().__init__(num_envs, seed)
        self.single_observation_space = Box(0.0, 1.0, (1,))
        self.single_action_space = Discrete(2)

    def _reset_rows(self, mask):
        pass

    def _obs(self):
        return np.zeros((self.num_envs, 1), dtype=np.float32)

    def _step_all(self, actions):
        reward = np.where(actions.reshape(-1) == 0, 1.0, -1.0).astype(np.float32)
        return reward, np.ones(self.num_envs, dtype=bool), None


class PolicyEnv(BatchedVecEnv):
    """One-hot obs in {e_0, e_1}; reward +1 iff action == obs index."""

    max_episode_steps = 1

    def __init__(self, num_envs: int = 1, seed: Optional[int] = None):
        super().__init__(num_envs, seed)
        self.single_observation_space = Box(0.0, 1.0, (2,))
        self.single_action_space = Discrete(2)
        self.idx = np.zeros(self.num_envs, dtype=np.int64)

    def _reset_rows(self, mask):
        self.idx[mask] = self.rng.integers(0, 2, int(mask.sum()))

    def _obs(self):
        obs = np.zeros((self.num_envs, 2), dtype=np.float32)
        obs[np.arange(self.num_envs), self.idx] = 1.0
        return obs

    def _step_all(self, actions):
        reward = np.where(actions.reshape(-1) == self.idx, 1.0, -1.0).astype(np.float32)
        return reward, np.ones(self.num_envs, dtype=bool), None


class ConstantRewardContActionsEnv(BatchedVecEnv):
    """Continuous actions, always reward 1, one step. Q* = 1."""

    max_episode_steps = 1
    q_values = np.array([[1.0]])

    def __init__(self, num_envs: int = 1, seed: Optional[int] = None):
        super().__init__(num_envs, seed)
        self.single_observation_space = Box(0.0, 1.0, (1,))
        self.single_action_space = Box(-1.0, 1.0, (1,))

    def _reset_rows(self, mask):
        pass

    def _obs(self):
        return np.zeros((self.num_envs, 1), dtype=np.float32)

    def _step_all(self, actions):
        return (
            np.ones(self.num_envs, dtype=np.float32),
            np.ones(self.num_envs, dtype=bool),
            None,
        )


class FixedObsPolicyContActionsEnv(BatchedVecEnv):
    """Fixed obs; reward = -(action - 0.5)^2.  Optimal action 0.5."""

    max_episode_steps = 1
    target_action = 0.5

    def __init__(self, num_envs: int = 1, seed: Optional[int] = None):
        super().__init__(num_envs, seed)
        self.single_observation_space = Box(0.0, 1.0, (1,))
        self.single_action_space = Box(-1.0, 1.0, (1,))

    def _reset_rows(self, mask):
        pass

    def _obs(self):
        return np.zeros((self.num_envs, 1), dtype=np.float32)

    def _step_all(self, actions):
        a = np.asarray(actions, dtype=np.float64).reshape(self.num_envs, -1)[:, 0]
        reward = -((a - self.target_action) ** 2)
        return reward.astype(np.float32), np.ones(self.num_envs, dtype=bool), None


# ---------------------------------------------------------------------------
# Runner helpers (reference parity: agilerl/utils/probe_envs.py check_* :1040+)
# Train briefly on a probe env, then assert the learned values match the
# env's analytic q/v/policy values.  Raise AssertionError on failure so they
# can be used both from tests and from user sanity-check scripts.
# ---------------------------------------------------------------------------


def _fill_probe_buffer(env, memory, steps: int, continuous: bool = False):
    import numpy as _np

    obs, _ = env.reset()
    for _ in range(steps):
        if continuous:
            action = _np.random.uniform(-1, 1, (env.num_envs,) + env.action_space.shape)
        else:
            action = _np.random.randint(0, env.action_space.n, env.num_envs)
        next_obs, reward, term, trunc, _ = env.step(action)
        memory.add(obs=obs, action=action, reward=reward, next_obs=next_obs,
                   done=term.astype("float32"))
        obs = next_obs


def check_q_learning_with_probe_env(env, algo_class, algo_args, learn_steps: int = 400,
                                    memory_size: int = 2000, atol: float = 0.15):
    """Off-policy discrete-action check: learned Q ≈ env.q_values."""
    import numpy as _np
    import torch as _torch

    from ..components.replay_buffer import ReplayBuffer

    agent = algo_class(env.observation_space, env.action_space, **algo_args)
    memory = ReplayBuffer(memory_size)
    _fill_probe_buffer(env, memory, min(memory_size // env.num_envs, 300))
    for _ in range(learn_steps):
        agent.learn(memory.sample(agent.batch_size))
    states = _np.eye(env.observation_space.shape[0], dtype=_np.float32) \
        if env.q_values.shape[0] > 1 else _np.zeros((1,) + env.observation_space.shape, _np.float32)
    with _torch.no_grad():
        q = agent.actor(_torch.as_tensor(states)).cpu().numpy()
    if q.ndim == 3:  # distributional (Rainbow): expectation already taken by actor? guard
        q = q.mean(-1)
    assert _np.allclose(q, env.q_values, atol=atol), f"Q {q} != {env.q_values}"
    return agent


def check_policy_q_learning_with_probe_env(env, algo_class, algo_args,
                                           learn_steps: int = 400,
                                           memory_size: int = 2000, atol: float = 0.2):
    """Off-policy continuous-action check (DDPG/TD3): critic ≈ env.q_values."""
    import numpy as _np
    import torch as _torch

    from ..components.replay_buffer import ReplayBuffer

    agent = algo_class(env.observation_space, env.action_space, **algo_args)
    memory = ReplayBuffer(memory_size)
    _fill_probe_buffer(env, memory, min(memory_size // env.num_envs, 300), continuous=True)
    for _ in range(learn_steps):
        agent.learn(memory.sample(agent.batch_size))
    with _torch.no_grad():
        obs = _torch.zeros((1,) + env.observation_space.shape)
        act = agent.actor(obs)
        critic = getattr(agent, "critic", None) or getattr(agent, "critic_1")
        q = critic(critic.preprocess(obs), act).cpu().numpy()
    assert _np.allclose(q, env.q_values, atol=atol), f"Q {q} != {env.q_values}"
    return agent


def check_on_policy_with_probe_env(env, algo_class, algo_args, rollouts: int = 20,
                                   n_steps: int = 32, target_action: int = None):
    """On-policy check (PPO): deterministic policy picks the rewarded action."""
    import torch as _torch

    from ..components.rollout_buffer import RolloutBuffer
    from ..rollouts.on_policy import collect_rollouts

    agent = algo_class(env.observation_space, env.action_space,
                       learn_step=n_steps, **algo_args)
    buf = RolloutBuffer(n_steps, env.num_envs, gamma=agent.gamma,
                        gae_lambda=agent.gae_lambda)
    obs = done = None
    for _ in range(rollouts):
        obs, done, _ = collect_rollouts(agent, env, buf, n_steps, obs, done)
        agent.learn(buf)
    if target_action is None:
        target_action = int(env.q_values.argmax(-1)[0])
    with _torch.no_grad():
        probe_obs = _torch.zeros((1,) + env.observation_space.shape)
        act = agent.actor.deterministic_action(agent.actor.preprocess(probe_obs))
    assert int(act.reshape(-1)[0]) == target_action, f"policy chose {act}, want {target_action}"
    return agent


def check_llm_policy_with_probe_env(agent, env, iterations: int = 15):
    """GRPO-family check on the token copy-task: mean reward must improve."""
    import numpy as _np

    from ..llm_envs import make_grpo_experiences

    rewards_hist = []
    for _ in range(iterations):
        prompts = env.reset()
        seqs = agent.get_action(prompts)
        rewards = env.score(seqs)
        rewards_hist.append(float(_np.mean(rewards)))
        agent.learn(make_grpo_experiences(env, seqs, rewards))
    first = _np.mean(rewards_hist[:3])
    last = _np.mean(rewards_hist[-5:])
    assert last > first, f"reward did not improve: {rewards_hist}"
    return rewards_hist


# reference utils/probe_envs.py spelling
check_policy_on_policy_with_probe_env = check_on_policy_with_probe_env
