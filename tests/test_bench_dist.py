"""bench.py distributed launch path (torchrun, gloo, CPU) — the exact
shape of the driver's round-end SCALE invocation."""
import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(420)
@pytest.mark.parametrize("decomp", ["coarse", "medium"])
def test_bench_torchrun_world2(decomp, tmp_path):
    env = dict(os.environ, MASTER_ADDR="127.0.0.1")
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--standalone",
         "--nnodes=1", "--nproc-per-node", "2", "--local-addr", "127.0.0.1",
         "bench.py", "--gpus", "2", "--steps", "1", "--warmup", "1",
         "--device", "cpu", "--config", "small", "--decomp", decomp],
        capture_output=True, text=True, cwd=ROOT, env=env, timeout=400)
    assert r.returncode == 0, r.stderr[-800:]
    lines = [l for l in r.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, r.stdout
    j = json.loads(lines[0])
    assert j["n_gpus"] == 2
    assert j["scaling"] == ("weak" if decomp == "coarse" else "strong")
