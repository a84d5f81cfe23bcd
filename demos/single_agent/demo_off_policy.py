"""DQN on CartPole with evolutionary HPO — the canonical off-policy demo.

Reference parity: demos/single_agent/demo_off_policy.py (same workflow,
written against the agilerl_amd API).  Runs on CPU in ~a minute; pass
--device cuda:0 on an MI355X box.
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__)))))

from agilerl_amd.algorithms import DQN
from agilerl_amd.components import ReplayBuffer
from agilerl_amd.envs import CartPoleVecEnv
from agilerl_amd.hpo import Mutations, TournamentSelection
from agilerl_amd.training import train_off_policy


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--device", default="cpu")
    p.add_argument("--max-steps", type=int, default=20_000)
    p.add_argument("--pop-size", type=int, default=4)
    args = p.parse_args()

    env = CartPoleVecEnv(num_envs=16, seed=42)
    pop = DQN.population(
        args.pop_size, env.observation_space, env.action_space,
        net_config={"arch": "mlp", "hidden_size": [64, 64]},
        batch_size=64, lr=1e-3, device=args.device,
    )
    memory = ReplayBuffer(50_000)
    tournament = TournamentSelection(tournament_size=2, elitism=True)
    mutations = Mutations(
        no_mutation=0.4, architecture=0.2, parameters=0.2,
        activation=0.0, rl_hp=0.2, rand_seed=42,
    )
    agents, _ = train_off_policy(
        env, "CartPole", "DQN", pop, memory,
        max_steps=args.max_steps, evo_steps=2_000, eval_loop=1,
        target=195.0, tournament=tournament, mutation=mutations,
    )
    best = max(agents, key=lambda a: a.fitness[-1])
    print(f"best fitness: {best.fitness[-1]:.1f}")
    best.save_checkpoint("dqn_cartpole_best.pt")


if __name__ == "__main__":
    main()
