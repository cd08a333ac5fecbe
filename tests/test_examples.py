"""Every example script must run clean on CPU (they auto-fall-back to the
tiny model / stub backends). Catches doc rot: examples are the first
thing a migrating user copies."""

import os
import subprocess
import sys

import pytest

EXAMPLES = [
    "basic_chat.py",
    "engine_generate.py",
    "agent_task.py",
    "memory_workflow.py",
    "federation_demo.py",
    "ask_with_search.py",
    "mcp_integration.py",
    "http_api.py",
    "paged_serving.py",
    "speculative_decode.py",
    "status_reporting.py",
]

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.parametrize("script", EXAMPLES)
def test_example_runs(script, tmp_path):
    env = dict(os.environ, MEMDIR_BASE=str(tmp_path / "Memdir"),
               HOME=str(tmp_path))
    r = subprocess.run(
        [sys.executable, os.path.join(ROOT, "examples", script)],
        capture_output=True, text=True, timeout=420, env=env)
    assert r.returncode == 0, f"{script} failed:\n{r.stderr[-1500:]}"
