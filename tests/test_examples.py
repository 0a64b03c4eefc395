"""The examples must keep running (they are user-facing documentation)."""
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(300)
def test_quickstart_runs():
    r = subprocess.run([sys.executable, "examples/quickstart.py"],
                       capture_output=True, text=True, cwd=ROOT, timeout=280)
    assert r.returncode == 0, r.stderr[-500:]
    assert "fit =" in r.stdout


@pytest.mark.timeout(420)
def test_distributed_example_runs():
    env = dict(os.environ, MASTER_ADDR="127.0.0.1")
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--standalone",
         "--nnodes=1", "--nproc-per-node", "2", "--local-addr", "127.0.0.1",
         "examples/distributed.py"],
        capture_output=True, text=True, cwd=ROOT, env=env, timeout=400)
    assert r.returncode == 0, r.stderr[-500:]
    assert "fit =" in r.stdout
