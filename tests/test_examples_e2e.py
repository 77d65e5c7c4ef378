"""Notebook-style E2E harness (the reference runs its 29
notebooks through nbtest/DatabricksUtilities; here every examples/*.py runs
as a subprocess in CI — same role: the documented end-to-end workflows must
actually execute)."""
import os
import subprocess
import sys

import pytest

EXAMPLES_DIR = os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "examples")

# 07 needs torchrun (multi-process); it gets its own wrapper below
SCRIPTS = sorted(f for f in os.listdir(EXAMPLES_DIR)
                 if f.endswith(".py") and not f.startswith("07"))


@pytest.mark.parametrize("script", SCRIPTS)
@pytest.mark.timeout(420)
def test_example_runs(script):
    env = dict(os.environ, MPLBACKEND="Agg",
               MMLSPARK_AMD_EXAMPLE_FAST="1")
    r = subprocess.run([sys.executable, os.path.join(EXAMPLES_DIR, script)],
                       capture_output=True, text=True, timeout=400, env=env)
    assert r.returncode == 0, (script, r.stdout[-1500:], r.stderr[-1500:])


@pytest.mark.timeout(420)
def test_example_distributed_torchrun():
    """07_distributed_training via torch.distributed.run, 2 ranks, gloo."""
    env = dict(os.environ, MASTER_ADDR="127.0.0.1",
               MMLSPARK_AMD_EXAMPLE_FAST="1")
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29951",
         os.path.join(EXAMPLES_DIR, "07_distributed_training.py")],
        capture_output=True, text=True, timeout=400, env=env)
    assert r.returncode == 0, (r.stdout[-1500:], r.stderr[-1500:])
