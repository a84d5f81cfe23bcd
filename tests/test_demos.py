"""Every shipped demo script runs end-to-end at reduced scale (slow set)."""

import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

DEMOS = [
    ("demos/single_agent/demo_off_policy.py", ["--max-steps", "2000"]),
    ("demos/single_agent/demo_on_policy.py", ["--max-steps", "4000"]),
    ("demos/single_agent/demo_curriculum.py", ["--max-steps", "1500"]),
    ("demos/multi_agent/demo_multi_agent.py", ["--max-steps", "2000"]),
    ("demos/bandits/demo_bandit.py", ["--max-steps", "400"]),
    ("demos/llm/demo_llm_finetuning.py", ["--iterations", "2"]),
    ("demos/llm/demo_multiturn.py", ["--iterations", "2"]),
]


@pytest.mark.slow
@pytest.mark.parametrize("script,args", DEMOS, ids=[d[0].split("/")[-1] for d in DEMOS])
def test_demo_runs(script, args, tmp_path):
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, script), *args],
        cwd=str(tmp_path), capture_output=True, text=True, timeout=420,
    )
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
