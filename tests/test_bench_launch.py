"""Launch-shape tests for the driver's bench contract.

The driver runs ``python -m torch.distributed.run --nnodes=1
--nproc-per-node N --master-addr 127.0.0.1 --master-port P bench.py``.
These tests run the EXACT same launch at N=2 on CPU (gloo) with a tiny
config — including evolution rounds (steps > EVO_EVERY) and therefore the
flat-tensor winner-weight broadcast — and validate the emitted JSON line.
"""

import json
import os
import socket
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BENCH = os.path.join(REPO, "bench.py")


def _free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _run_bench(extra, nproc=2, timeout=420):
    cmd = [
        sys.executable, "-m", "torch.distributed.run",
        "--nnodes=1", f"--nproc-per-node={nproc}",
        "--master-addr", "127.0.0.1", "--master-port", str(_free_port()),
        BENCH, "--gpus", str(nproc),
    ] + extra
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    out = subprocess.run(
        cmd, capture_output=True, text=True, timeout=timeout, cwd=REPO, env=env
    )
    assert out.returncode == 0, f"bench failed:\n{out.stdout[-3000:]}\n{out.stderr[-3000:]}"
    lines = [l for l in out.stdout.splitlines() if l.strip().startswith("{")]
    assert lines, f"no JSON emitted:\n{out.stdout[-2000:]}"
    return json.loads(lines[-1])


@pytest.mark.slow
def test_bench_ppo_n2_gloo_with_evolution():
    """Headline workload at N=2 under gloo: 5 steps crosses EVO_EVERY=4, so
    tournament plan broadcast + flat-tensor parent transfer + mutations all
    run inside the timed region — the exact 8-GPU code path, CPU-sized."""
    result = _run_bench([
        "--steps", "5", "--warmup", "1",
        "--num-envs", "16", "--learn-step", "16", "--no-graph",
    ])
    assert result["metric"] == "env_steps_per_sec"
    assert result["n_gpus"] == 2
    assert result["steps"] == 5
    assert result["value"] > 0
    assert result["config"]["pop_size"] == 8
    # whole-job aggregate: 8 agents x 16 envs x 16 steps x 5 bench steps
    expected_steps = 8 * 16 * 16 * 5
    measured = result["value"] * (result["ms_per_step"] * 5 / 1000.0)
    assert abs(measured - expected_steps) / expected_steps < 0.05


@pytest.mark.slow
def test_bench_grpo_tiny_n2_gloo():
    result = _run_bench([
        "--workload", "grpo", "--model-size", "tiny",
        "--steps", "2", "--warmup", "1", "--seq-len", "64", "--grpo-batch", "8",
    ])
    assert result["metric"] == "train_tokens_per_sec"
    assert result["n_gpus"] == 2
    assert result["value"] > 0
    assert result["config"]["global_batch"] == 16


@pytest.mark.slow
def test_bench_rainbow_n2_gloo():
    result = _run_bench([
        "--workload", "rainbow", "--steps", "2", "--warmup", "1",
        "--num-envs", "8",
    ])
    assert result["metric"] == "env_steps_per_sec"
    assert result["n_gpus"] == 2
    assert result["value"] > 0
    assert result["config"]["pop_size"] == 4


@pytest.mark.slow
def test_bench_maddpg_n2_gloo():
    result = _run_bench([
        "--workload", "maddpg", "--steps", "2", "--warmup", "1",
        "--num-envs", "8",
    ])
    assert result["metric"] == "env_steps_per_sec"
    assert result["n_gpus"] == 2
    assert result["config"]["pop_size"] == 8
    assert result["value"] > 0
