"""Smoke-run every entry script (the reference's user surface: one file per
algorithm with all settings inline, run directly)."""
import os
import subprocess
import sys
from pathlib import Path

import pytest

ROOT = Path(__file__).resolve().parents[1]
EXAMPLES = ["grpo", "ppo", "rloo", "remax", "raft", "reinforce", "grpo_r1"]


@pytest.mark.parametrize("name", EXAMPLES)
def test_example_runs(name, tmp_path):
    env = dict(os.environ, NANORLHF_TEST_TMP=str(tmp_path))
    r = subprocess.run([sys.executable, str(ROOT / "examples" / f"{name}.py")],
                       capture_output=True, text=True, timeout=600, env=env,
                       cwd=str(tmp_path))
    assert r.returncode == 0, (r.stdout[-1500:], r.stderr[-1500:])
