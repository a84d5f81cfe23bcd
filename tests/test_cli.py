"""CLI entry-point tests (subprocess)."""

import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_train_cli(tmp_path):
    manifest = tmp_path / "m.yaml"
    manifest.write_text(
        """
algorithm:
  name: DQN
  hyperparameters: {batch_size: 32, lr: 0.001}
environment: {env_id: CartPole-v1, num_envs: 4}
network:
  arch: mlp
  encoder_config: {hidden_size: [16]}
training: {max_steps: 300, pop_size: 1, evo_steps: 150, eval_loop: 1}
"""
    )
    out = subprocess.run(
        [sys.executable, "-m", "agilerl_amd.train", str(manifest), "--device", "cpu",
         "--csv", str(tmp_path / "log.csv")],
        cwd=ROOT, capture_output=True, text=True, timeout=300,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    assert (tmp_path / "log.csv").exists()


def test_arena_cli(tmp_path):
    manifest = tmp_path / "m.yaml"
    manifest.write_text(
        """
algorithm:
  name: DQN
  hyperparameters: {batch_size: 16}
environment: {env_id: CartPole-v1, num_envs: 2}
training: {max_steps: 100, pop_size: 1, evo_steps: 50, eval_loop: 1}
"""
    )
    ws = str(tmp_path / "arena")
    out = subprocess.run(
        [sys.executable, "-m", "agilerl_amd.arena.cli", "--workspace", ws,
         "submit", str(manifest)],
        cwd=ROOT, capture_output=True, text=True, timeout=300,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    assert "completed" in out.stdout
    out2 = subprocess.run(
        [sys.executable, "-m", "agilerl_amd.arena.cli", "--workspace", ws, "list"],
        cwd=ROOT, capture_output=True, text=True, timeout=60,
    )
    assert out2.stdout.strip().startswith("exp-")


def test_bench_contract_cpu():
    """bench.py emits exactly one JSON line with the driver-contract keys."""
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
         "--num-envs", "8", "--pop-size", "2"],
        cwd=ROOT, capture_output=True, text=True, timeout=600,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1
    payload = json.loads(lines[0])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in payload, key


@pytest.mark.slow
@pytest.mark.parametrize("workload", ["dqn", "rainbow", "maddpg"])
def test_bench_other_workloads_cpu(workload):
    """Every BASELINE workload's bench path emits a valid contract line on CPU."""
    out = subprocess.run(
        [sys.executable, "bench.py", "--workload", workload, "--steps", "2",
         "--warmup", "1", "--num-envs", "8", "--pop-size", "2"],
        cwd=ROOT, capture_output=True, text=True, timeout=600,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1
    payload = json.loads(lines[0])
    assert payload["value"] > 0
    assert payload["config"]


@pytest.mark.slow
def test_bench_grpo_tiny_cpu():
    out = subprocess.run(
        [sys.executable, "bench.py", "--workload", "grpo", "--model-size", "tiny",
         "--steps", "2", "--warmup", "1", "--seq-len", "64", "--grpo-batch", "4"],
        cwd=ROOT, capture_output=True, text=True, timeout=900,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    payload = json.loads(lines[0])
    assert payload["unit"].startswith("tokens")


def test_kernel_isa_report_tool():
    """The committed ISA table can be regenerated from the shipped .so."""
    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, "tools", "kernel_isa_report.py")],
        capture_output=True, text=True, timeout=120,
    )
    assert out.returncode == 0, out.stderr[-1000:]
    assert "rmsnorm_fwd_kernel" in out.stdout
    assert "swiglu_fwd_kernel" in out.stdout


@pytest.mark.slow
def test_bench_distributed_two_ranks_cpu():
    """Exactly the driver's multi-GPU launch shape, on CPU gloo: torchrun
    2 ranks, whole-job aggregate from rank 0 only."""
    env = dict(os.environ)
    env.update({"MASTER_ADDR": "127.0.0.1"})
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29517", "bench.py", "--gpus", "2",
         "--steps", "2", "--warmup", "1", "--num-envs", "8", "--pop-size", "2"],
        cwd=ROOT, capture_output=True, text=True, timeout=900, env=env,
    )
    assert out.returncode == 0, out.stderr[-3000:]
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout[-2000:]
    payload = json.loads(lines[0])
    assert payload["n_gpus"] == 2
    assert payload["value"] > 0


class TestReferenceManifests:
    """Every reference training YAML must validate against our manifest
    schema (north star: 'configs work unchanged')."""

    REF_DIR = "/root/reference/configs/training"

    def test_all_reference_yamls_validate(self):
        import glob

        import yaml as _yaml

        from agilerl_amd.models.manifest import TrainingManifest, resolve_algo_class

        files = sorted(glob.glob(f"{self.REF_DIR}/**/*.yaml", recursive=True))
        if not files:
            pytest.skip("reference configs not present in this image")
        failures = []
        for f in files:
            try:
                doc = _yaml.safe_load(open(f))
                manifest = TrainingManifest.model_validate(doc)
                resolve_algo_class(manifest.algorithm.name)
                manifest.env_spec()  # `name:`-keyed env sections must resolve
            except Exception as e:  # noqa: BLE001 - collect all failures
                failures.append(f"{f}: {e}")
        assert not failures, "\n".join(failures)

    def test_unknown_algorithm_is_explicit(self):
        from agilerl_amd.models.manifest import TrainingManifest

        with pytest.raises(Exception, match="Unknown algorithm"):
            TrainingManifest.model_validate({
                "algorithm": {"name": "NotARealAlgo"},
                "environment": {"name": "CartPole-v1"},
            })


class TestReferenceConfigEndToEnd:
    """Run actual reference YAMLs through LocalTrainer (tiny overrides on
    the budget fields only) — validation alone doesn't prove the configs
    WORK here."""

    @pytest.mark.slow
    @pytest.mark.parametrize("rel", [
        "dqn/dqn.yaml", "ppo/ppo.yaml", "dqn/dqn_rainbow.yaml",
        "td3.yaml", "ddpg/ddpg.yaml", "multi_agent/maddpg.yaml",
        "cqn.yaml", "dqn/dqn_mfpbt.yaml", "ppo/ppo_recurrent.yaml",
        "multi_agent/matd3.yaml", "multi_agent/ippo.yaml",
        "multi_agent/maddpg_mfpbt.yaml", "multi_agent/ippo_pong.yaml",
        "ppo/ppo_mfpbt.yaml", "cqn_mfpbt.yaml",
        "multi_input.yaml", "ppo/ppo_image.yaml",
    ])
    def test_reference_yaml_trains(self, rel):
        import yaml as _yaml

        from agilerl_amd.models.manifest import TrainingManifest
        from agilerl_amd.training.trainer import LocalTrainer

        path = f"/root/reference/configs/training/{rel}"
        if not os.path.exists(path):
            pytest.skip("reference configs not present")
        doc = _yaml.safe_load(open(path))
        manifest = TrainingManifest.model_validate(doc)
        # shrink ONLY the budget knobs; algorithm/network/mutation sections
        # stay exactly as the reference wrote them
        manifest.training.max_steps = 600
        manifest.training.evo_steps = 256
        manifest.training.pop_size = 2
        manifest.environment["num_envs"] = 4
        trainer = LocalTrainer(manifest, device="cpu")
        results = trainer.train()
        assert results is not None


class TestReferenceFieldAliases:
    def test_reference_mutation_and_target_fields_apply(self):
        """Reference spellings (no_mut/arch_mut/.../new_layer/target_score)
        must MAP, not silently fall back to defaults."""
        import yaml as _yaml

        from agilerl_amd.models.manifest import TrainingManifest

        doc = _yaml.safe_load(open("/root/reference/configs/training/dqn/dqn.yaml"))
        m = TrainingManifest.model_validate(doc)
        p = m.mutation.probabilities
        assert (p.no_mutation, p.architecture, p.parameters, p.activation,
                p.rl_hp) == (0.4, 0.2, 0.2, 0.2, 0.2)
        assert m.mutation.new_layer_prob == 0.2
        assert m.mutation.mutation_sd == 0.1
        assert m.training.target == 200.0
        # rl_hp_selection bounds flow too
        assert m.mutation.rl_hp_selection["lr"]["max"] == 0.01


class TestReferenceBanditConfig:
    def test_reference_neural_ucb_yaml_with_reference_csvs(self):
        """The reference bandit manifest + the reference's own IRIS CSVs
        train through the labelled-dataset BanditEnv."""
        import yaml as _yaml

        from agilerl_amd.models.manifest import TrainingManifest
        from agilerl_amd.training.trainer import LocalTrainer

        path = "/root/reference/configs/training/bandit/neural_ucb.yaml"
        feats = "/root/reference/tests/data/iris_features.csv"
        if not (os.path.exists(path) and os.path.exists(feats)):
            pytest.skip("reference assets absent")
        doc = _yaml.safe_load(open(path))
        doc["environment"]["features"] = feats
        doc["environment"]["targets"] = "/root/reference/tests/data/iris_targets.csv"
        doc["training"].update({"max_steps": 200, "pop_size": 2, "evo_steps": 100})
        m = TrainingManifest.model_validate(doc)
        out = LocalTrainer(m, device="cpu").train()
        assert out is not None


class TestUtilsParityFunctions:
    """Reference utils.py:275/:308/:1588/:1651 convenience functions."""

    def test_calculate_vectorized_scores(self):
        import numpy as np

        from agilerl_amd.utils.utils import calculate_vectorized_scores

        r = np.array([[1, 1, 1, 1], [2, 2, 2, 2]], dtype=float)
        t = np.array([[0, 1, 0, 1], [0, 0, 0, 0]])
        assert calculate_vectorized_scores(r, t) == [2.0, 8.0]
        assert calculate_vectorized_scores(r, t, only_first_episode=False) == [2.0, 2.0, 8.0]

    def test_make_skill_vect_envs_reward_hook(self):
        from agilerl_amd.utils.utils import make_skill_vect_envs
        from agilerl_amd.wrappers.learning import Skill

        class Doubler(Skill):
            def skill_reward(self, obs, reward, terminated, truncated, info):
                return reward * 2, terminated, truncated

        env = make_skill_vect_envs("CartPole-v1", Doubler, num_envs=2, seed=0)
        env.reset()
        import numpy as np

        _, reward, _, _, _ = env.step(np.zeros(2, dtype=np.int64))
        assert (reward == 2.0).all()

    def test_print_hyperparams_runs(self, capsys):
        from agilerl_amd.algorithms.dqn import DQN
        from agilerl_amd.spaces import Box, Discrete
        from agilerl_amd.utils.utils import print_hyperparams

        print_hyperparams([DQN(Box(-1, 1, (4,)), Discrete(2))])
        out = capsys.readouterr().out
        assert "Agent ID: 0" in out and "lr" in out
