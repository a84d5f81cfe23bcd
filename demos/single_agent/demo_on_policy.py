"""PPO on LunarLander with evolutionary HPO (the BASELINE headline config).

Reference parity: demos/single_agent/demo_on_policy.py.  On an MI355X box
the torch GPU env + hipGraph capture path activates automatically; this is
the same code path bench.py measures.
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__)))))

from agilerl_amd.algorithms import PPO
from agilerl_amd.envs import LunarLanderVecEnv
from agilerl_amd.hpo import Mutations, TournamentSelection
from agilerl_amd.training import train_on_policy


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--device", default="cpu")
    p.add_argument("--max-steps", type=int, default=100_000)
    p.add_argument("--pop-size", type=int, default=4)
    args = p.parse_args()

    env = LunarLanderVecEnv(num_envs=16, seed=42)
    pop = PPO.population(
        args.pop_size, env.observation_space, env.action_space,
        net_config={"arch": "mlp", "hidden_size": [64, 64]},
        batch_size=256, learn_step=128, lr=3e-4, device=args.device,
    )
    agents, _ = train_on_policy(
        env, "LunarLander", "PPO", pop,
        max_steps=args.max_steps, evo_steps=10_000, eval_loop=1, target=200.0,
        tournament=TournamentSelection(tournament_size=2, elitism=True),
        mutation=Mutations(no_mutation=0.4, architecture=0.2, parameters=0.2,
                           activation=0.0, rl_hp=0.2, rand_seed=42),
    )
    best = max(agents, key=lambda a: a.fitness[-1])
    print(f"best fitness: {best.fitness[-1]:.1f}")


if __name__ == "__main__":
    main()
