"""Randomized mutation-sequence stress (the arch sweep's big sibling).

Reference analog: tests/test_hpo/test_mutation.py's breadth.  For a grid
of (algorithm, encoder arch), run many RANDOM evolution rounds through
the real Mutations class with every category enabled, learning after
each round, then require clone + checkpoint round-trips to reproduce the
final mutated agent exactly.  This is the suite that caught the
multi-input replay desync and the ModuleDict delegation bug.
"""

import numpy as np
import pytest
import torch

from agilerl_amd.hpo import Mutations

ROUNDS = 8


def _box_batch(space, n=8):
    import numpy as _np

    return torch.as_tensor(_np.stack([space.sample() for _ in range(n)])).float()


def _make(algo, arch):
    from agilerl_amd.spaces import Box, Discrete

    torch.manual_seed(0)
    if arch == "cnn":
        obs = Box(0, 255, (1, 10, 10))
        cfg = {"arch": "cnn", "channel_size": [4, 4], "kernel_size": [3, 3],
               "stride_size": [1, 1]}
    else:
        obs = Box(-1, 1, (6,))
        cfg = {"arch": "mlp", "hidden_size": [16, 16]}
    if algo == "DQN":
        from agilerl_amd.algorithms.dqn import DQN

        return DQN(obs, Discrete(3), net_config=cfg, batch_size=8), obs, "q"
    if algo == "Rainbow":
        from agilerl_amd.algorithms.dqn_rainbow import RainbowDQN

        return RainbowDQN(obs, Discrete(3), net_config=cfg, batch_size=8,
                          n_step=1), obs, "rainbow"
    if algo == "PPO":
        from agilerl_amd.algorithms.ppo import PPO

        return PPO(obs, Discrete(3), net_config=cfg, batch_size=32,
                   update_epochs=1, learn_step=8), obs, "ppo"
    from agilerl_amd.algorithms.ddpg import DDPG

    return DDPG(obs, Box(-1, 1, (2,)), net_config=cfg, batch_size=8), obs, "ddpg"


def _learn_once(agent, kind, obs_space):
    if kind == "ppo":
        n = 32
        flat = {
            "obs": _box_batch(obs_space, n),
            "action": torch.randint(0, 3, (n,)),
            "log_prob": torch.randn(n) * 0.1,
            "advantages": torch.randn(n),
            "returns": torch.randn(n),
            "value": torch.randn(n),
            "done": torch.zeros(n),
            "reward": torch.zeros(n),
        }
        stats = agent.learn(flat)
        assert all(np.isfinite(v) for v in stats.values())
        return
    batch = {
        "obs": _box_batch(obs_space),
        "action": (
            torch.rand(8, 2) * 2 - 1 if kind == "ddpg"
            else torch.randint(0, 3, (8, 1))
        ),
        "reward": torch.randn(8, 1),
        "next_obs": _box_batch(obs_space),
        "done": torch.zeros(8, 1),
    }
    if kind == "rainbow":
        batch["weights"] = torch.ones(8)
        batch["idxs"] = torch.arange(8)
    out = agent.learn(batch)
    vals = out if isinstance(out, tuple) else (out,)
    assert all(np.isfinite(float(v)) for v in vals if v is not None)


@pytest.mark.slow
@pytest.mark.parametrize("algo", ["DQN", "Rainbow", "PPO", "DDPG"])
@pytest.mark.parametrize("arch", ["mlp", "cnn"])
def test_random_mutation_rounds_then_roundtrip(algo, arch, tmp_path):
    agent, obs_space, kind = _make(algo, arch)
    muts = Mutations(
        no_mutation=0.1, architecture=0.45, parameters=0.15,
        activation=0.15, rl_hp=0.15, rand_seed=1234,
    )
    applied = []
    for _ in range(ROUNDS):
        muts.mutation([agent])
        applied.append(agent.mut)
        _learn_once(agent, kind, obs_space)

    # something must actually have mutated across the rounds
    assert any(m not in ("None", None) for m in applied), applied

    # clone reproduces the final mutated agent exactly
    clone = agent.clone(9)
    x = _box_batch(obs_space, 4)
    pol = "actor"
    torch.testing.assert_close(
        getattr(clone, pol)(x), getattr(agent, pol)(x)
    )

    # checkpoint round-trip reproduces it too (full save/load cycle)
    path = tmp_path / "fuzz.pt"
    agent.save_checkpoint(str(path))
    from agilerl_amd.algorithms.core.base import EvolvableAlgorithm

    restored = EvolvableAlgorithm.load(str(path))
    torch.testing.assert_close(
        getattr(restored, pol)(x), getattr(agent, pol)(x)
    )
    # and the restored agent keeps learning
    _learn_once(restored, kind, obs_space)


@pytest.mark.slow
@pytest.mark.parametrize("algo", ["MADDPG", "IPPO"])
def test_multi_agent_random_mutation_rounds(algo, tmp_path):
    """MA agents (ModuleDict nets) through random evolution rounds — the
    r1 ModuleDict delegation bug class."""
    from agilerl_amd.spaces import Box, Discrete

    torch.manual_seed(0)
    ids = ["a0", "a1"]
    obs = {k: Box(-1, 1, (5,)) for k in ids}
    if algo == "MADDPG":
        from agilerl_amd.algorithms.maddpg import MADDPG

        act = {k: Box(-1, 1, (2,)) for k in ids}
        agent = MADDPG(obs, act, agent_ids=ids, batch_size=8,
                       net_config={"arch": "mlp", "hidden_size": [16]})
    else:
        from agilerl_amd.algorithms.ippo import IPPO

        act = {k: Discrete(3) for k in ids}
        agent = IPPO(obs, act, agent_ids=ids, batch_size=16,
                     net_config={"arch": "mlp", "hidden_size": [16]},
                     update_epochs=1, learn_step=8)
    muts = Mutations(no_mutation=0.1, architecture=0.5, parameters=0.2,
                     activation=0.2, rl_hp=0.0, rand_seed=7)

    def ma_obs(n=8):
        return {k: torch.randn(n, 5) for k in ids}

    mutated = 0
    for _ in range(6):
        muts.mutation([agent])
        mutated += agent.mut not in ("None", None)
        if algo == "MADDPG":
            batch = {
                "obs": ma_obs(), "action": {k: torch.rand(8, 2) * 2 - 1 for k in ids},
                "reward": {k: torch.randn(8, 1) for k in ids},
                "next_obs": ma_obs(), "done": {k: torch.zeros(8, 1) for k in ids},
            }
            out = agent.learn(batch)
            for v in (out.values() if isinstance(out, dict) else [out]):
                vals = v if isinstance(v, tuple) else (v,)
                assert all(np.isfinite(float(x)) for x in vals if x is not None)
        else:
            # IPPO: real per-agent rollout buffers (its learn interface)
            from agilerl_amd.components.rollout_buffer import RolloutBuffer

            buffers = {}
            for k in ids:
                buf = RolloutBuffer(capacity=8, num_envs=2)
                for _ in range(8):
                    buf.add(
                        obs=torch.randn(2, 5),
                        action=torch.randint(0, 3, (2,)),
                        reward=torch.randn(2),
                        done=torch.zeros(2),
                        value=torch.randn(2),
                        log_prob=torch.randn(2) * 0.1,
                    )
                buf.compute_returns_and_advantages(torch.randn(2))
                buffers[k] = buf
            stats = agent.learn(buffers)
            assert all(np.isfinite(v) for v in stats.values())
    assert mutated > 0

    clone = agent.clone(3)
    x = ma_obs(4)
    for k in ids:
        pol = agent.actors[k] if hasattr(agent, "actors") else agent.actor[k]
        pol_c = clone.actors[k] if hasattr(clone, "actors") else clone.actor[k]
        torch.testing.assert_close(pol_c(x[k]), pol(x[k]))
