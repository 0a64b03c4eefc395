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


@pytest.mark.timeout(420)
def test_cli_cpd_torchrun_world2(tmp_path):
    """`torchrun -m splatt_amd cpd` — the mpirun-splatt-cpd analog — must
    run end to end on CPU/gloo and write rank-0 factor files."""
    import torch  # noqa: F401  (ensures torch importable before subprocess)
    import splatt_amd as sp
    t = sp.SpTensor.synthetic([60, 45, 80], 4000, seed=21).fixed()
    tns = tmp_path / "t.tns"
    t.save(tns)
    env = dict(os.environ, MASTER_ADDR="127.0.0.1",
               PYTHONPATH=ROOT + os.pathsep + os.environ.get("PYTHONPATH", ""))
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--standalone",
         "--nnodes=1", "--nproc-per-node", "2", "--local-addr", "127.0.0.1",
         "-m", "splatt_amd", "cpd", str(tns), "-r", "6", "-i", "4", "-t", "0",
         "--device", "cpu"],
        capture_output=True, text=True, cwd=str(tmp_path), env=env,
        timeout=400)
    assert r.returncode == 0, r.stderr[-800:]
    assert "Final fit:" in r.stdout
    for m in range(3):
        assert (tmp_path / f"mode{m + 1}.mat").exists()


@pytest.mark.timeout(600)
def test_bench_driver_argv_world8(tmp_path):
    """The driver's EXACT SCALE argv at N=8 (no extra flags) over gloo with
    the RCCL reduce-scatter/all-gather primitives forced on — end-to-end
    coverage of the default medium-grain strong-scaling path the 8-GPU
    run executes. Only the tensor size is overridden (env),
    never the code path."""
    env = dict(os.environ, MASTER_ADDR="127.0.0.1",
               SPLATT_BENCH_CONFIG="small",
               SPLATT_FORCE_RS_PRIMS="1",
               SPLATT_COMM_CHUNK_MIN_MB="0")
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--standalone",
         "--nnodes=1", "--nproc-per-node", "8", "--local-addr", "127.0.0.1",
         "bench.py", "--gpus", "8", "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, cwd=ROOT, env=env, timeout=580)
    assert r.returncode == 0, r.stderr[-1500:]
    lines = [l for l in r.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, r.stdout
    j = json.loads(lines[0])
    assert j["n_gpus"] == 8
    assert j["scaling"] == "strong"
    assert "medium-grid" in j["config"]["parallelism"]
    assert 0.0 <= j["config"]["fit"] < 1.0


@pytest.mark.timeout(600)
def test_bench_driver_argv_world8_4mode(tmp_path):
    """Config 5's SHAPE class (4-mode, medium-grain, N=8) through the
    driver's exact argv over gloo + forced RS/AG primitives."""
    env = dict(os.environ, MASTER_ADDR="127.0.0.1",
               SPLATT_BENCH_CONFIG="small4",
               SPLATT_FORCE_RS_PRIMS="1",
               SPLATT_COMM_CHUNK_MIN_MB="0")
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--standalone",
         "--nnodes=1", "--nproc-per-node", "8", "--local-addr", "127.0.0.1",
         "bench.py", "--gpus", "8", "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, cwd=ROOT, env=env, timeout=580)
    assert r.returncode == 0, r.stderr[-1500:]
    j = json.loads([l for l in r.stdout.strip().splitlines()
                    if l.startswith("{")][0])
    assert j["n_gpus"] == 8 and j["scaling"] == "strong"
    assert len(j["config"]["dims"]) == 4
