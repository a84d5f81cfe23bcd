"""Composite golden-path test: several features chained end to end —
custom actor network + RSNorm wrapper + training loop with CSV logging +
elite checkpoint + serving the checkpoint over HTTP."""

import json
import os

import numpy as np
import pytest
import torch


def test_golden_path(tmp_path):
    from agilerl_amd.algorithms import DQN
    from agilerl_amd.components import ReplayBuffer
    from agilerl_amd.envs import CartPoleVecEnv
    from agilerl_amd.hpo import Mutations, TournamentSelection
    from agilerl_amd.logger import CSVLogger
    from agilerl_amd.training import train_off_policy
    from agilerl_amd.wrappers import MakeEvolvable, RSNorm

    torch.manual_seed(0), np.random.seed(0)
    env = CartPoleVecEnv(num_envs=8, seed=0)
    user_net = torch.nn.Sequential(
        torch.nn.Linear(4, 32), torch.nn.ReLU(), torch.nn.Linear(32, 2)
    )
    pop = [
        RSNorm(DQN(env.observation_space, env.action_space,
                   actor_network=MakeEvolvable(user_net, torch.randn(1, 4)),
                   batch_size=32, lr=1e-3, index=0)),
        RSNorm(DQN(env.observation_space, env.action_space,
                   net_config={"arch": "mlp", "hidden_size": [32]},
                   batch_size=32, lr=1e-3, index=1)),
    ]
    csv_path = str(tmp_path / "log.csv")
    elite_path = str(tmp_path / "elite.pt")
    agents, hist = train_off_policy(
        env, "CartPole", "DQN", pop, ReplayBuffer(5000),
        max_steps=3000, evo_steps=1000, eval_loop=1,
        tournament=TournamentSelection(2, True),
        mutation=Mutations(no_mutation=0.5, architecture=0.2, parameters=0.1,
                           activation=0.0, rl_hp=0.2, rand_seed=0),
        save_elite=True, elite_path=elite_path,
        loggers=[CSVLogger(csv_path)], verbose=False,
    )
    assert len(hist) >= 1
    assert os.path.exists(csv_path) and os.path.getsize(csv_path) > 0
    assert os.path.exists(elite_path)

    # serve the elite checkpoint
    fastapi = pytest.importorskip("fastapi")
    from fastapi.testclient import TestClient

    from agilerl_amd.serve import create_app, load_agent

    agent = load_agent(elite_path)
    client = TestClient(create_app(agent))
    r = client.post("/predict", json={"obs": np.zeros((2, 4)).tolist()})
    assert r.status_code == 200
    assert len(r.json()["action"]) == 2
