#!/usr/bin/env python3
"""Curriculum learning with Skill wrappers (reference docs' skills tutorial,
wrappers/learning.py Skill).

Trains a DQN CartPole agent through two skills in sequence — "center"
(bonus for staying near x=0) then the raw balancing task — carrying the
same agent (and its replay buffer) across stages.  Each stage wraps the
vectorized env in a Skill subclass whose ``skill_reward`` reshapes the
batched rewards.
"""

from __future__ import annotations

import argparse
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__)))))

from agilerl_amd.algorithms.dqn import DQN
from agilerl_amd.components import ReplayBuffer
from agilerl_amd.envs.registry import make_vect_envs
from agilerl_amd.wrappers.learning import Skill


class CenterSkill(Skill):
    """Reward shaping: bonus for keeping the cart near the track center."""

    def skill_reward(self, obs, reward, terminated, truncated, info):
        bonus = 0.5 * (1.0 - np.minimum(np.abs(np.asarray(obs)[..., 0]) / 2.4, 1.0))
        return reward + bonus, terminated, truncated


def run_stage(env, agent, buf, steps: int, eps: float = 0.2) -> float:
    obs, _ = env.reset(seed=0)
    returns, acc = [], np.zeros(env.num_envs)
    for it in range(steps):
        if np.random.rand() < eps:
            action = np.array([env.single_action_space.sample()
                               for _ in range(env.num_envs)])
        else:
            action = agent.get_action(obs, epsilon=eps)
        next_obs, reward, term, trunc, _ = env.step(action)
        buf.add(obs=obs, action=action, reward=reward, next_obs=next_obs,
                done=term.astype(np.float32))
        acc += np.asarray(reward)
        done = term | trunc
        if done.any():
            returns.extend(acc[done].tolist())
            acc[done] = 0.0
        obs = next_obs
        if len(buf) >= agent.batch_size and it % agent.learn_step == 0:
            agent.learn(buf.sample(agent.batch_size))
    return float(np.mean(returns[-20:])) if returns else 0.0


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--max-steps", type=int, default=6000)
    p.add_argument("--num-envs", type=int, default=8)
    args = p.parse_args()

    base = make_vect_envs("CartPole-v1", num_envs=args.num_envs, seed=0)
    agent = DQN(base.observation_space, base.action_space, lr=1e-3, batch_size=64,
                net_config={"arch": "mlp", "hidden_size": [64]})
    buf = ReplayBuffer(50_000)

    stage_steps = args.max_steps // 2
    skill_env = CenterSkill(base)
    r1 = run_stage(skill_env, agent, buf, stage_steps)
    print(f"stage 1 (center skill): mean return {r1:.1f}")

    r2 = run_stage(base, agent, buf, stage_steps)
    print(f"stage 2 (full task):    mean return {r2:.1f}")
    print("curriculum demo done")


if __name__ == "__main__":
    main()
