"""Arena client + multi-agent probe tests."""

import os

import numpy as np
import pytest

from agilerl_amd.arena import ArenaClient, ArenaError


class TestArenaClient:
    def _manifest(self):
        return {
            "algorithm": {"name": "DQN", "hyperparameters": {"batch_size": 32, "lr": 1e-3}},
            "environment": {"env_id": "CartPole-v1", "num_envs": 4},
            "network": {"arch": "mlp", "encoder_config": {"hidden_size": [16]}},
            "training": {"max_steps": 300, "pop_size": 1, "evo_steps": 150, "eval_loop": 1},
        }

    def test_requires_login(self, tmp_path):
        client = ArenaClient(workspace=str(tmp_path))
        with pytest.raises(ArenaError, match="not logged in"):
            client.submit_experiment(self._manifest(), run=False)

    def test_submit_and_status(self, tmp_path):
        client = ArenaClient(workspace=str(tmp_path))
        client.login()
        handle = client.submit_experiment(self._manifest(), run=True)
        assert handle.status == "completed"
        assert client.experiment_status(handle.experiment_id)["status"] == "completed"
        assert handle.experiment_id in client.list_experiments()
        assert os.path.exists(os.path.join(handle.workspace, "manifest.yaml"))

    def test_validate_environment(self, tmp_path):
        client = ArenaClient(workspace=str(tmp_path))
        report = client.validate_environment({"env_id": "CartPole-v1", "algorithm": "DQN"})
        assert report["valid"]
        report2 = client.validate_environment({"env_id": "NoSuchEnv", "algorithm": "DQN"})
        assert report2["warnings"]

    def test_resume(self, tmp_path):
        client = ArenaClient(workspace=str(tmp_path))
        client.login()
        handle = client.submit_experiment(self._manifest(), run=False)
        resumed = client.resume_experiment(handle.experiment_id)
        assert resumed.status == "completed"


class TestArenaTrainer:
    """Reference trainer.py:872 ArenaTrainer: manifest -> submit via client."""

    def _manifest(self):
        return {
            "algorithm": {"name": "DQN", "hyperparameters": {"batch_size": 32, "lr": 1e-3}},
            "environment": {"env_id": "CartPole-v1", "num_envs": 4},
            "network": {"arch": "mlp", "encoder_config": {"hidden_size": [16]}},
            "training": {"max_steps": 300, "pop_size": 1, "evo_steps": 150, "eval_loop": 1},
        }

    def test_train_submits_and_completes(self, tmp_path):
        from agilerl_amd.training import ArenaTrainer

        client = ArenaClient(workspace=str(tmp_path))
        client.login()
        trainer = ArenaTrainer.from_manifest(self._manifest(), client=client)
        handle = trainer.train()
        assert handle.status == "completed"
        assert handle.experiment_id in trainer.list_experiments()
        # resume re-runs from the stored manifest
        resumed = trainer.resume_from_checkpoint(handle.experiment_id, max_steps=300)
        assert resumed.status == "completed"

    def test_rejects_mfpbt(self, tmp_path):
        from agilerl_amd.training import ArenaTrainer

        m = self._manifest()
        m["selection_strategy"] = {"strategy": "multi_frequency",
                                   "evolution_frequency_ratios": [1, 2]}
        with pytest.raises(ValueError, match="MF-PBT"):
            ArenaTrainer.from_manifest(m, client=ArenaClient(workspace=str(tmp_path)))


class TestArenaDatasetsAndDeploy:
    def test_dataset_roundtrip(self, tmp_path):
        import numpy as np

        client = ArenaClient(workspace=str(tmp_path / "ws"))
        client.login()
        ds_file = tmp_path / "transitions.npz"
        np.savez(ds_file, observations=np.zeros((4, 2), dtype=np.float32))
        ds_id = client.upload_dataset(str(ds_file))
        assert ds_id in client.list_datasets()
        stored = client.dataset_path(ds_id)
        with np.load(stored) as z:
            assert z["observations"].shape == (4, 2)
        with pytest.raises(ArenaError):
            client.dataset_path("ds-missing")

    def test_deploy_and_predict(self, tmp_path):
        import numpy as np

        from agilerl_amd.algorithms import DQN
        from agilerl_amd.spaces import Box, Discrete

        client = ArenaClient(workspace=str(tmp_path))
        client.login()
        exp_dir = tmp_path / "exp-abc"
        exp_dir.mkdir()
        agent = DQN(Box(-1.0, 1.0, (4,)), Discrete(2),
                    net_config={"arch": "mlp", "hidden_size": [16]})
        agent.save_checkpoint(str(exp_dir / "ckpt_final.pt"))
        dep = client.deploy("exp-abc")
        assert dep.info()["algo"] == "DQN"
        actions = dep.predict(np.zeros((3, 4), dtype=np.float32))
        assert len(actions) == 3
        with pytest.raises(ArenaError):
            client.deploy("exp-nope")


class TestMAProbes:
    def test_joint_action_env(self):
        from agilerl_amd.envs.probe_ma import JointActionMAEnv

        env = JointActionMAEnv(num_envs=4)
        obs, _ = env.reset()
        actions = {"agent_0": np.zeros(4, dtype=int), "agent_1": np.zeros(4, dtype=int)}
        _, rewards, term, trunc, _ = env.step(actions)
        np.testing.assert_allclose(rewards["agent_0"], 1.0)
        actions["agent_1"] = np.ones(4, dtype=int)
        env.reset()
        _, rewards, _, _, _ = env.step(actions)
        np.testing.assert_allclose(rewards["agent_0"], -1.0)

    def test_maddpg_learns_joint_q(self):
        """Centralized critic must learn Q(s, a0=0, a1=0)=1 vs mixed=-1."""
        import torch

        from agilerl_amd.algorithms import MADDPG
        from agilerl_amd.components import ReplayBuffer
        from agilerl_amd.envs.probe_ma import JointActionMAEnv

        torch.manual_seed(0)
        np.random.seed(0)
        env = JointActionMAEnv(num_envs=8, seed=0)
        agent = MADDPG(env.observation_spaces, env.action_spaces, agent_ids=env.agents,
                       batch_size=64, lr_critic=1e-2, tau=0.1,
                       net_config={"arch": "mlp", "hidden_size": [32]})
        buf = ReplayBuffer(2000)
        obs, _ = env.reset()
        for _ in range(150):
            acts = {a: np.random.randint(0, 2, 8) for a in env.agents}
            raw = {a: np.eye(2, dtype=np.float32)[acts[a]] for a in env.agents}
            next_obs, rewards, term, trunc, _ = env.step(acts)
            buf.add(obs=obs, action=raw, reward={a: rewards[a] for a in env.agents},
                    next_obs=next_obs, done={a: term[a].astype(np.float32) for a in env.agents})
            obs = next_obs
        for _ in range(300):
            agent.learn(buf.sample(64))
        joint_obs = torch.zeros(1, 2)
        both_zero = torch.tensor([[1.0, 0.0, 1.0, 0.0]])
        mixed = torch.tensor([[1.0, 0.0, 0.0, 1.0]])
        with torch.no_grad():
            q_good = float(agent.critics["agent_0"](joint_obs, both_zero))
            q_bad = float(agent.critics["agent_0"](joint_obs, mixed))
        assert q_good > 0.5
        assert q_bad < 0.0
