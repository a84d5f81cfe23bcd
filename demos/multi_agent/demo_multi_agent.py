"""MADDPG on the speaker/listener MPE task with evolutionary HPO.

Reference parity: demos/multi_agent/demo_multi_agent.py (the reference uses
PettingZoo's simple_speaker_listener; here the first-party vectorized MPE
port — same observation/action contract).
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__)))))

from agilerl_amd.algorithms import MADDPG
from agilerl_amd.components import ReplayBuffer
from agilerl_amd.envs.mpe import SpeakerListenerVecEnv
from agilerl_amd.hpo import Mutations, TournamentSelection
from agilerl_amd.training import train_multi_agent_off_policy


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--device", default="cpu")
    p.add_argument("--max-steps", type=int, default=40_000)
    p.add_argument("--pop-size", type=int, default=4)
    args = p.parse_args()

    env = SpeakerListenerVecEnv(num_envs=8, seed=42)
    pop = MADDPG.population(
        args.pop_size, env.observation_spaces, env.action_spaces,
        agent_ids=env.agents,
        net_config={"arch": "mlp", "hidden_size": [64, 64]},
        batch_size=128, device=args.device,
    )
    memory = ReplayBuffer(100_000)
    agents, _ = train_multi_agent_off_policy(
        env, "speaker_listener", "MADDPG", pop, memory,
        max_steps=args.max_steps, evo_steps=5_000, eval_loop=1,
        tournament=TournamentSelection(tournament_size=2, elitism=True),
        mutation=Mutations(no_mutation=0.4, architecture=0.2, parameters=0.2,
                           activation=0.0, rl_hp=0.2, rand_seed=42),
    )
    best = max(agents, key=lambda a: a.fitness[-1])
    print(f"best fitness: {best.fitness[-1]:.2f}")


if __name__ == "__main__":
    main()
