#!/usr/bin/env python3
"""Wall-clock-to-target-reward measurement (BASELINE metric 2).

PPO pop=N evolutionary HPO on the first-party LunarLander dynamics;
stops when any agent's fitness (5-rollout mean episode return) reaches
the target.  Uses bench.py's exact runner, so evolution (tournament +
mutations + graph recapture) is inside the measured wall-clock.

Scope note: the env is this repo's Box2D-free LunarLander
reimplementation — identical observation/action spaces and reward
structure to LunarLander-v2 (potential-based shaping + terminal ±100),
but first-party rigid-body dynamics, numpy<->torch parity-tested
(tests/test_torch_envs.py).  The solve threshold is therefore a claim
about THESE dynamics, not about Gymnasium's Box2D implementation
(no Box2D/gymnasium exists in the offline image to cross-validate).
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import bench  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--pop-size", type=int, default=8)
    p.add_argument("--num-envs", type=int, default=None)
    p.add_argument("--learn-step", type=int, default=bench.LEARN_STEP)
    p.add_argument("--target", type=float, default=200.0)
    p.add_argument("--max-seconds", type=float, default=600.0)
    p.add_argument("--no-graph", action="store_true")
    args = p.parse_args()
    import torch

    if args.num_envs is None:
        args.num_envs = 65536 if torch.cuda.is_available() else 64
    args.workload = "ppo"
    args.steps = 0
    args.warmup = 0

    runner = bench.BenchRunner(args)
    t0 = time.perf_counter()
    steps_done = 0
    best = -float("inf")
    while True:
        steps_done += runner.bench_step()
        best = max(
            best,
            max((a.fitness[-1] for a in runner.pop.agents.values() if a.fitness),
                default=-float("inf")),
        )
        elapsed = time.perf_counter() - t0
        print(f"[{elapsed:7.1f}s] cycle {runner.step_count:3d} "
              f"best_fitness {best:8.2f} env_steps {steps_done:.2e}", flush=True)
        if best >= args.target or elapsed > args.max_seconds:
            break
    elapsed = time.perf_counter() - t0
    print(json.dumps({
        "metric": "wall_clock_to_target_reward",
        "solved": best >= args.target,
        "seconds": elapsed,
        "best_fitness": best,
        "target": args.target,
        "pop_size": args.pop_size,
        "num_envs_per_agent": args.num_envs,
        "env_steps": steps_done,
        "evo_rounds": runner.step_count // bench.EVO_EVERY,
        "env": "first-party LunarLander dynamics (Box2D-free; see module docstring)",
    }), flush=True)


if __name__ == "__main__":
    main()
