"""CLI + auxiliary subsystems: reorder, graph/hypergraph, tiling, timers,
bench harness (reference tests: reorder_test.c, graph.c, tile_*_test.c)."""
import torch

import splatt_amd as sp
from splatt_amd import reorder as ro
from splatt_amd.cli import main as cli_main
from splatt_amd.graph import (graph_mpartite, graph_write, hgraph_nnz,
                              hgraph_uncut, hgraph_write, part_read)
from splatt_amd.tile import densetile, next_tileid_in_layer, tile_coords, tile_id


def dense_of(t):
    d = torch.zeros(*t.dims, dtype=torch.float64)
    d.index_put_(tuple(t.inds), t.vals.double(), accumulate=True)
    return d


def test_perm_roundtrip(small3):
    perm = ro.perm_rand(small3.dims, seed=3)
    assert perm.is_valid()
    t2 = ro.perm_apply(small3, perm)
    # permuted dense tensor == dense tensor indexed by perms
    a = dense_of(small3)
    b = dense_of(t2)
    assert torch.equal(a[perm.perms[0]][:, perm.perms[1]][:, :, perm.perms[2]], b)


def test_perm_bfs_valid(small3):
    perm = ro.perm_bfs(small3)
    assert perm.is_valid()


def test_perm_hgraph_valid(small3):
    part = torch.randint(0, 4, (small3.nnz,))
    perm = ro.perm_hgraph(small3, part)
    assert perm.is_valid()


def test_graphs(small3, tmp_path):
    g = graph_mpartite(small3)
    assert g.nvtxs == sum(small3.dims)
    assert int(g.adj_ptr[-1]) == g.adj_ind.numel()
    # symmetric: total weight even
    assert int(g.adj_wgt.sum()) % 2 == 0
    graph_write(g, tmp_path / "g.graph")
    hg = hgraph_nnz(small3)
    assert hg.nvtxs == small3.nnz
    assert hg.eind.numel() == small3.nnz * small3.nmodes
    hgraph_write(hg, tmp_path / "h.hgraph")
    part = torch.zeros(small3.nnz, dtype=torch.int64)
    assert hgraph_uncut(hg, part).numel() == hg.nhedges


def test_densetile(small3):
    tiled, tiling = densetile(small3, [4, 4, 4])
    assert tiled.nnz == small3.nnz
    assert int(tiling.tile_ptr[-1]) == small3.nnz
    assert abs(float(tiled.vals.sum() - small3.vals.sum())) < 1e-9
    # tile id math round-trips
    for tid in range(64):
        assert tile_id(tile_coords(tid, [4, 4, 4]), [4, 4, 4]) == tid
    # layer traversal covers each layer exactly once (incl. prime dims)
    dims = [3, 5, 2]
    seen = set()
    for layer in range(5):
        tid = -1
        while True:
            tid = next_tileid_in_layer(tid, dims, 1, layer)
            if tid < 0:
                break
            assert tid not in seen
            seen.add(tid)
    assert len(seen) == 3 * 5 * 2


def test_cli_roundtrip(tmp_path, small3, capsys):
    tns = str(tmp_path / "t.tns")
    small3.save(tns)
    assert cli_main(["stats", tns]) == 0
    assert cli_main(["check", tns, "--fix", str(tmp_path / "f.tns"),
                     "--compress"]) == 0
    assert cli_main(["convert", tns, str(tmp_path / "t.bin"), "-t", "bin"]) == 0
    assert cli_main(["reorder", tns, str(tmp_path / "r.tns"),
                     "--type", "rand"]) == 0
    t2 = sp.load(str(tmp_path / "t.bin"))
    assert t2.nnz == small3.nnz
    import os
    os.chdir(tmp_path)
    assert cli_main(["cpd", tns, "-r", "4", "-i", "3", "--device", "cpu",
                     "--native"]) == 0
    assert (tmp_path / "mode1.mat").exists()
    assert (tmp_path / "lambda.mat").exists()


def test_cli_bench(tmp_path):
    t = sp.SpTensor.synthetic([30, 25, 40], 3000, seed=5)
    tns = str(tmp_path / "b.tns")
    t.save(tns)
    assert cli_main(["bench", tns, "-r", "8", "-N", "1", "--device", "cpu",
                     "-a", "csf,stream", "--validate"]) == 0


def test_timers():
    from splatt_amd.utils.timers import TimerRegistry
    reg = TimerRegistry()
    with reg.time("X"):
        pass
    with reg.time("X"):
        pass
    assert reg.timers["X"].count == 2
    assert "X" in reg.report()


def test_ccp_partition():
    """Optimality properties over varied weight shapes (reference
    thread_partition_test.c: unit/random/sorted/fibonacci x many parts)."""
    from splatt_amd._ext import native
    import random
    rng = random.Random(7)
    shapes = {
        "unit": [1] * 53,
        "random": [rng.randint(1, 100) for _ in range(97)],
        "sorted": sorted(rng.randint(1, 50) for _ in range(64)),
        "fib": [1, 1, 2, 3, 5, 8, 13, 21, 34, 55, 89, 144],
        "spike": [1] * 30 + [1000] + [1] * 30,
    }
    for name, w in shapes.items():
        for parts in (1, 2, 3, 7, 13, 31):
            bounds, bn = native().partition_weighted(w, parts)
            assert bounds[0] == 0 and bounds[-1] == len(w)
            assert all(a <= b for a, b in zip(bounds, bounds[1:]))
            # bottleneck matches the heaviest part
            loads = [sum(w[a:b]) for a, b in zip(bounds, bounds[1:])]
            assert max(loads) == bn, (name, parts)
            # optimal: no feasible split with a smaller bottleneck
            assert bn >= max(w)
            assert bn >= (sum(w) + parts - 1) // parts
            _, bn2 = native().partition_weighted(w, parts + 1)
            assert bn2 <= bn


def test_legacy_bench_algs():
    import splatt_amd as sp
    from splatt_amd.benchmarks import bench_mttkrp
    t = sp.SpTensor.synthetic([25, 20, 30], 2000, seed=5)
    res = bench_mttkrp(t, 8, ["csf", "giga", "ttbox"], 1, device="cpu",
                       validate=True)
    for alg in ("csf", "giga", "ttbox"):
        assert res[alg]["validated"], alg


def test_bench_contract_json(tmp_path):
    """bench.py must emit exactly one valid JSON line with the driver's
    contract fields (CPU small config)."""
    import json
    import os
    import subprocess
    import sys as _sys
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [_sys.executable, "bench.py", "--device", "cpu", "--config", "small",
         "--steps", "1", "--warmup", "1"],
        capture_output=True, text=True, cwd=root, timeout=300)
    assert r.returncode == 0, r.stderr[-500:]
    lines = [l for l in r.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1
    j = json.loads(lines[0])
    for field in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                  "ms_per_step", "higher_is_better", "scaling",
                  "vs_baseline", "dtype", "data", "config"):
        assert field in j, field
    assert j["n_gpus"] == 1 and j["steps"] == 1
    assert j["higher_is_better"] is True


def test_cli_reorder_partfile(tmp_path, small3):
    """graph/hgraph reorder driven by an external partition file."""
    tns = str(tmp_path / "t.tns")
    small3.save(tns)
    nv = sum(small3.dims)
    part = tmp_path / "parts.txt"
    part.write_text("\n".join(str(i % 4) for i in range(nv)))
    out = str(tmp_path / "r.tns")
    assert cli_main(["reorder", tns, out, "--type", "graph",
                     "--partfile", str(part),
                     "--permfile", str(tmp_path / "p")]) == 0
    t2 = sp.load(out)
    assert t2.nnz == small3.nnz
    # perm files round-trip
    from splatt_amd import reorder as ro
    perm = ro.perm_read(str(tmp_path / "p"), 3)
    assert perm.is_valid()
    # hgraph variant: nnz-partition file
    part2 = tmp_path / "nnzparts.txt"
    part2.write_text("\n".join(str(i % 3) for i in range(small3.nnz)))
    assert cli_main(["reorder", tns, str(tmp_path / "r2.tns"),
                     "--type", "hgraph", "--partfile", str(part2)]) == 0


def test_hgraph_fib_and_perm_matrix(small3):
    from splatt_amd.graph import hgraph_fib
    hg = hgraph_fib(small3, mode=2)
    assert hg.nvtxs > 0
    assert int(hg.eptr[-1]) == hg.eind.numel()
    # every pin is a valid fiber id
    assert int(hg.eind.max()) < hg.nvtxs
    perm = ro.perm_rand(small3.dims, seed=1)
    A = torch.rand(small3.dims[0], 4, dtype=torch.float64)
    B = ro.perm_matrix(A, perm.perms[0])
    assert torch.equal(B[perm.iperms[0]], A)


def test_cli_cpd_deterministic_flag(tmp_path, capsys):
    """--deterministic sets the env knob and forces ALLMODE; on CPU the
    run itself is the (already deterministic) host path."""
    import os
    import splatt_amd as sp
    from splatt_amd.cli import main
    t = sp.SpTensor.synthetic([12, 10, 14], 300, seed=5)
    f = tmp_path / "t.tns"
    t.save(f)
    rc = main(["cpd", str(f), "-r", "4", "-i", "2", "--device", "cpu",
               "--deterministic", "--nowrite"])
    assert rc == 0
    assert os.environ.pop("SPLATT_DETERMINISTIC", None) == "1"


def test_cli_convert_csr_and_fib_hgraph(tmp_path):
    import splatt_amd as sp
    from splatt_amd.cli import main
    t = sp.SpTensor.synthetic([6, 5, 4], 40, seed=8).fixed()
    f = tmp_path / "t.tns"
    t.save(f)
    out = tmp_path / "t.csr"
    assert main(["convert", str(f), str(out), "-t", "csr", "-m", "1"]) == 0
    lines = out.read_text().strip().splitlines()
    nr, nc, nnz = map(int, lines[0].split())
    assert (nr, nc) == (5, 24) and nnz == t.nnz
    assert len(lines) == 1 + nr
    out2 = tmp_path / "t.fib.hg"
    assert main(["convert", str(f), str(out2), "-t", "fib_hgraph"]) == 0
    assert out2.exists()


def test_cli_stats_hparts(tmp_path, capsys):
    import torch
    import splatt_amd as sp
    from splatt_amd.cli import main
    t = sp.SpTensor.synthetic([10, 8, 12], 200, seed=9).fixed()
    f = tmp_path / "t.tns"
    t.save(f)
    part = (torch.arange(t.nnz) * 3 // t.nnz)    # 3 contiguous parts
    pf = tmp_path / "t.part"
    pf.write_text("\n".join(str(int(x)) for x in part) + "\n")
    assert main(["stats", str(f), "--part", str(pf)]) == 0
    out = capsys.readouterr().out
    assert "NPARTS=3" in out and "cut slices" in out


def test_cli_bench_thread_sweep(tmp_path, capsys):
    import splatt_amd as sp
    from splatt_amd.cli import main
    t = sp.SpTensor.synthetic([15, 12, 18], 600, seed=2).fixed()
    f = tmp_path / "t.tns"
    t.save(f)
    rc = main(["bench", str(f), "-r", "4", "-a", "flat,stream", "-N", "1",
               "--device", "cpu", "--threads", "1,2", "--validate"])
    assert rc == 0
    out = capsys.readouterr().out
    assert "flat@t1" in out and "flat@t2" in out and "stream" in out
    assert "MISMATCH" not in out


def test_comm_stats_touched_vs_exchanged():
    """comm_stats with a CsfSet reports rows actually touched by local
    nonzeros vs whole-chunk exchanged rows."""
    import splatt_amd as sp
    from splatt_amd.parallel.grid import GridDecomp, comm_stats
    t = sp.SpTensor.synthetic([40, 30, 50], 500, seed=8).fixed(dedup=True)
    dec = GridDecomp.create(list(t.dims))          # world 1
    cs = sp.csf_alloc(t, "all")
    st = comm_stats(dec, t.nnz, 8, cs=cs)
    assert len(st["touched_rows_per_mode"]) == 3
    for m in range(3):
        uniq = int(t.inds[m].unique().numel())
        assert st["touched_rows_per_mode"][m] == uniq
        assert st["exchanged_rows_per_mode"][m] == 0   # no comm at world 1
