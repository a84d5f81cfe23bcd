"""Profiling harness: PPO collect+learn under torch.profiler / rocprofv3.

Reference parity: demos/single_agent/performance_flamegraph_*.py
(the reference ships torch.profiler/cProfile flamegraph demos; on MI355X
run this under `rocprofv3 --kernel-trace --stats -- python demos/profile_ppo.py`
for per-kernel time, or standalone for the torch.profiler ROCm trace).
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from torch.profiler import ProfilerActivity, profile

from agilerl_amd.algorithms.ppo import PPO
from agilerl_amd.components import RolloutBuffer
from agilerl_amd.rollouts.on_policy import collect_rollouts, collect_rollouts_device


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--num-envs", type=int, default=256)
    p.add_argument("--steps", type=int, default=4)
    p.add_argument("--trace-out", default=None, help="chrome trace path")
    args = p.parse_args()

    use_cuda = torch.cuda.is_available()
    device = "cuda:0" if use_cuda else "cpu"
    if use_cuda:
        from agilerl_amd.envs.torch_envs import LunarLanderTorchVecEnv

        env = LunarLanderTorchVecEnv(args.num_envs, device=device, seed=0)
        collect = collect_rollouts_device
    else:
        from agilerl_amd.envs import LunarLanderVecEnv

        env = LunarLanderVecEnv(args.num_envs, seed=0)
        collect = collect_rollouts

    agent = PPO(env.single_observation_space, env.single_action_space,
                learn_step=128, batch_size=4096, device=device)
    buf = RolloutBuffer(128, args.num_envs, device=device,
                        gamma=agent.gamma, gae_lambda=agent.gae_lambda)

    obs = done = None
    obs, done, _ = collect(agent, env, buf, 128, obs, done)  # warmup
    agent.learn(buf)

    activities = [ProfilerActivity.CPU]
    if use_cuda:
        activities.append(ProfilerActivity.CUDA)
    with profile(activities=activities, record_shapes=False) as prof:
        for _ in range(args.steps):
            obs, done, _ = collect(agent, env, buf, 128, obs, done)
            agent.learn(buf)
    print(prof.key_averages().table(
        sort_by="cuda_time_total" if use_cuda else "cpu_time_total", row_limit=25))
    if args.trace_out:
        prof.export_chrome_trace(args.trace_out)


if __name__ == "__main__":
    main()
