"""Manifest + LocalTrainer + CLI tests."""

import os

import numpy as np
import pytest
import yaml

from agilerl_amd.models import TrainingManifest, algo_workload
from agilerl_amd.training.trainer import LocalTrainer

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


class TestManifest:
    def test_example_manifests_validate(self):
        for rel in [
            "configs/training/dqn/dqn.yaml",
            "configs/training/ppo/ppo_lunarlander.yaml",
            "configs/training/maddpg/maddpg_speaker_listener.yaml",
            "configs/training/rainbow/rainbow.yaml",
        ]:
            m = TrainingManifest.from_yaml(os.path.join(ROOT, rel))
            assert m.algorithm.name
            assert m.env_spec() is not None

    def test_workload_mapping(self):
        assert algo_workload("DQN") == "off_policy"
        assert algo_workload("PPO") == "on_policy"
        assert algo_workload("MADDPG") == "multi_agent_off_policy"
        assert algo_workload("IPPO") == "multi_agent_on_policy"
        assert algo_workload("NeuralUCB") == "bandit"
        with pytest.raises(KeyError):
            algo_workload("NoSuchAlgo")

    def test_roundtrip(self, tmp_path):
        m = TrainingManifest.model_validate(
            {"algorithm": {"name": "DQN"}, "environment": {"env_id": "CartPole-v1"}}
        )
        p = str(tmp_path / "m.yaml")
        m.to_yaml(p)
        m2 = TrainingManifest.from_yaml(p)
        assert m2.algorithm.name == "DQN"

    def test_unknown_algo_rejected(self):
        with pytest.raises(Exception):
            TrainingManifest.model_validate({"algorithm": {"name": "Bogus"}})


class TestLocalTrainer:
    def _manifest(self, **training):
        t = dict(max_steps=2000, pop_size=2, evo_steps=500, eval_loop=1)
        t.update(training)
        return {
            "algorithm": {"name": "DQN", "hyperparameters": {"batch_size": 32, "lr": 1e-3}},
            "environment": {"env_id": "CartPole-v1", "num_envs": 4},
            "network": {"arch": "mlp", "encoder_config": {"hidden_size": [32]}},
            "training": t,
        }

    def test_dqn_end_to_end(self):
        trainer = LocalTrainer.from_manifest(self._manifest())
        agents, hist = trainer.train()
        assert len(agents) == 2
        assert len(hist) >= 1

    def test_ppo_end_to_end(self):
        m = {
            "algorithm": {"name": "PPO", "hyperparameters": {"batch_size": 64, "learn_step": 32}},
            "environment": {"env_id": "CartPole-v1", "num_envs": 4},
            "training": {"max_steps": 600, "pop_size": 2, "evo_steps": 128, "eval_loop": 1},
        }
        agents, hist = LocalTrainer.from_manifest(m).train()
        assert len(agents) == 2

    def test_maddpg_end_to_end(self):
        m = {
            "algorithm": {"name": "MADDPG", "hyperparameters": {"batch_size": 32}},
            "environment": {"type": "pettingzoo", "env_id": "simple_speaker_listener_v4", "num_envs": 4},
            "network": {"arch": "mlp", "encoder_config": {"hidden_size": [32]}},
            "training": {"max_steps": 300, "pop_size": 2, "evo_steps": 100, "eval_loop": 1},
        }
        agents, hist = LocalTrainer.from_manifest(m).train()
        assert len(agents) == 2

    def test_checkpointing(self, tmp_path):
        ckpt = str(tmp_path / "pop.pt")
        trainer = LocalTrainer.from_manifest(
            self._manifest(checkpoint=500, checkpoint_path=ckpt, max_steps=2200)
        )
        trainer.train()
        assert os.path.exists(str(tmp_path / "pop_0.pt"))

    def test_rainbow_per_manifest(self):
        m = {
            "algorithm": {"name": "RainbowDQN", "hyperparameters": {"batch_size": 32, "n_step": 2}},
            "environment": {"env_id": "CartPole-v1", "num_envs": 4},
            "replay_buffer": {"max_size": 5000, "per": True, "n_step": 2},
            "training": {"max_steps": 800, "pop_size": 2, "evo_steps": 300, "eval_loop": 1},
        }
        agents, _ = LocalTrainer.from_manifest(m).train()
        assert agents[0].algo == "RainbowDQN"


class TestBandits:
    def test_neural_ucb_learns_synthetic(self):
        from agilerl_amd.algorithms import NeuralUCB
        from agilerl_amd.envs.bandit import SyntheticBanditEnv
        from agilerl_amd.training.train_bandits import train_bandits

        env = SyntheticBanditEnv(context_dim=4, num_arms=3, seed=0)
        pop = NeuralUCB.population(
            1, env.observation_space, env.action_space,
            net_config={"arch": "mlp", "hidden_size": [32]}, lr=1e-2, batch_size=32,
        )
        agents, hist = train_bandits(
            env, "synth", "NeuralUCB", pop, max_steps=600, evo_steps=300,
            eval_steps=100, verbose=False,
        )
        assert np.isfinite(hist[-1][0])

    def test_neural_ts_action(self):
        from agilerl_amd.algorithms import NeuralTS
        from agilerl_amd.envs.bandit import SyntheticBanditEnv

        env = SyntheticBanditEnv(context_dim=4, num_arms=3, seed=0)
        agent = NeuralTS(env.observation_space, env.action_space,
                         net_config={"arch": "mlp", "hidden_size": [16]})
        ctx = env.reset()
        arm = agent.get_action(ctx)
        assert 0 <= arm < 3

    def test_bandit_env_from_data(self):
        from agilerl_amd.envs.bandit import BanditEnv

        X = np.random.rand(50, 5).astype(np.float32)
        y = np.random.randint(0, 3, 50)
        env = BanditEnv(X, y)
        ctx = env.reset()
        assert ctx.shape == (3, 15)
        r, ctx2 = env.step(0)
        assert r in (0.0, 1.0)

    def test_bandit_manifest(self):
        m = {
            "algorithm": {"name": "NeuralUCB", "hyperparameters": {"batch_size": 16, "lr": 1e-2}},
            "environment": {"type": "bandit", "context_dim": 4, "num_arms": 3},
            "training": {"max_steps": 200, "pop_size": 1, "evo_steps": 100, "eval_steps": 50},
        }
        agents, _ = LocalTrainer.from_manifest(m).train()
        assert agents[0].algo == "NeuralUCB"


class TestResume:
    def test_resume_from_population_checkpoint(self, tmp_path):
        ckpt = str(tmp_path / "pop.pt")
        m = {
            "algorithm": {"name": "DQN", "hyperparameters": {"batch_size": 32, "lr": 1e-3}},
            "environment": {"env_id": "CartPole-v1", "num_envs": 4},
            "network": {"arch": "mlp", "encoder_config": {"hidden_size": [16]}},
            "training": {"max_steps": 800, "pop_size": 2, "evo_steps": 400,
                         "eval_loop": 1, "checkpoint": 400, "checkpoint_path": ckpt},
        }
        agents1, _ = LocalTrainer.from_manifest(m).train()
        import os

        assert os.path.exists(str(tmp_path / "pop_0.pt"))
        # resume: new trainer picks up saved weights and step counts
        m2 = dict(m)
        m2["training"] = dict(m["training"], resume_from_checkpoint=ckpt, max_steps=1200)
        trainer2 = LocalTrainer.from_manifest(m2)
        env = trainer2._make_env()
        pop2 = trainer2._make_population(env)
        assert pop2[0].steps[-1] > 0  # carried over from the checkpoint


def test_rl_hp_selection_bounds_applied():
    """manifest mutation.rl_hp_selection overrides per-HP mutation ranges."""
    from agilerl_amd.training import LocalTrainer

    manifest = {
        "algorithm": {"name": "DQN", "hyperparameters": {"batch_size": 32}},
        "environment": {"type": "gym", "env_id": "CartPole-v1", "num_envs": 4},
        "network": {"arch": "mlp", "encoder_config": {"hidden_size": [16]}},
        "mutation": {"rl_hp_selection": {"lr": {"min": 1e-4, "max": 1e-3},
                                         "batch_size": {"min": 16, "max": 64}}},
        "training": {"max_steps": 100, "pop_size": 2, "evo_steps": 50},
    }
    trainer = LocalTrainer.from_manifest(manifest)
    env = trainer._make_env()
    pop = trainer._make_population(env)
    for agent in pop:
        lr = agent.hp_config.config["lr"]
        assert lr.min == pytest.approx(1e-4) and lr.max == pytest.approx(1e-3)
        bs = agent.hp_config.config["batch_size"]
        assert bs.min == 16 and bs.max == 64 and isinstance(bs.min, int)
