"""Profiling harness: recurrent (LSTM) PPO collect + BPTT learn.

Reference parity: demos/single_agent/performance_flamegraph_rnn_*.py /
performance_flamegraph_lunar_lander_rnn.py.  Run standalone for a
torch.profiler table, or under
`rocprofv3 --kernel-trace --stats -- python demos/profile_ppo_rnn.py`
for per-kernel time on an MI355X.
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from torch.profiler import ProfilerActivity, profile

from agilerl_amd.algorithms.ppo import PPO
from agilerl_amd.components import RolloutBuffer
from agilerl_amd.envs import LunarLanderVecEnv
from agilerl_amd.rollouts.on_policy import collect_rollouts_recurrent


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--num-envs", type=int, default=64)
    p.add_argument("--rollout", type=int, default=64)
    p.add_argument("--iters", type=int, default=3)
    p.add_argument("--trace-out", default=None)
    args = p.parse_args()

    device = "cuda:0" if torch.cuda.is_available() else "cpu"
    env = LunarLanderVecEnv(args.num_envs, seed=0)
    agent = PPO(env.observation_space, env.action_space, recurrent=True,
                net_config={"arch": "lstm", "hidden_state_size": 64},
                learn_step=args.rollout, batch_size=256, device=device)
    buf = RolloutBuffer(args.rollout, args.num_envs, gamma=agent.gamma,
                        gae_lambda=agent.gae_lambda, device=device)
    obs = done = hidden = None
    # warmup
    obs, done, hidden, _ = collect_rollouts_recurrent(
        agent, env, buf, args.rollout, obs, done, hidden)
    agent.learn(buf)

    activities = [ProfilerActivity.CPU]
    if device != "cpu":
        activities.append(ProfilerActivity.CUDA)
    with profile(activities=activities) as prof:
        for _ in range(args.iters):
            obs, done, hidden, _ = collect_rollouts_recurrent(
                agent, env, buf, args.rollout, obs, done, hidden)
            agent.learn(buf)
    key = "cuda_time_total" if device != "cpu" else "cpu_time_total"
    print(prof.key_averages().table(sort_by=key, row_limit=15))
    if args.trace_out:
        prof.export_chrome_trace(args.trace_out)


if __name__ == "__main__":
    main()
