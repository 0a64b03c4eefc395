"""Multi-process distributed CPD over gloo (CPU, world_size 2) — validates
the collective schedule that runs over RCCL on the GPU node. The key
property (reference design: rank-invariant mpi_mat_rand, mpi/mpi_io.c:1097):
fit at world N == fit at world 1 for the same seed."""
import os

import pytest
import torch
import torch.multiprocessing as mp

import splatt_amd as sp
from splatt_amd.parallel.dist_cpd import (
    build_shard_csf, dist_cpd_als, localize_shard, partition_rows)

DIMS = [40, 30, 60]
NNZ = 6000
RANK_F = 8
ITERS = 6
SEED = 77


def test_partition_rows():
    assert partition_rows(10, 4, 0) == (0, 3)
    assert partition_rows(10, 4, 1) == (3, 3)
    assert partition_rows(10, 4, 2) == (6, 2)
    assert partition_rows(10, 4, 3) == (8, 2)
    total = sum(partition_rows(97, 7, r)[1] for r in range(7))
    assert total == 97


def test_localize_shard():
    t = sp.SpTensor.synthetic(DIMS, 500, seed=3)
    row0, nloc = partition_rows(DIMS[2], 2, 1)
    s = localize_shard(t, 2, row0, nloc)
    assert s.dims[2] == nloc
    assert int(s.inds[2].max()) < nloc
    n0 = localize_shard(t, 2, 0, partition_rows(DIMS[2], 2, 0)[1]).nnz
    assert n0 + s.nnz == t.nnz


def _worker(rank, world, file_store, result_q):
    torch.distributed.init_process_group(
        "gloo", init_method=f"file://{file_store}", rank=rank,
        world_size=world)
    try:
        t = sp.SpTensor.synthetic(DIMS, NNZ, seed=SEED)
        part_mode = 2
        row0, nloc = partition_rows(DIMS[part_mode], world, rank)
        shard = localize_shard(t, part_mode, row0, nloc)
        cs = build_shard_csf(shard, list(DIMS), "two")
        opts = sp.CpdOptions(max_iters=ITERS, tolerance=0.0, seed=SEED)
        k = dist_cpd_als(cs, part_mode, row0, list(DIMS), RANK_F, opts)
        if rank == 0:
            result_q.put(("fit", k.fit, k.niters))
    finally:
        torch.distributed.destroy_process_group()


@pytest.mark.timeout(300)
def test_dist_fit_matches_single_process(tmp_path):
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    # single-process reference
    t = sp.SpTensor.synthetic(DIMS, NNZ, seed=SEED)
    opts = sp.CpdOptions(max_iters=ITERS, tolerance=0.0, seed=SEED)
    k1 = sp.cpd_als(t, RANK_F, opts)

    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    store = str(tmp_path / "store")
    procs = [ctx.Process(target=_worker, args=(r, 2, store, q))
             for r in range(2)]
    for p in procs:
        p.start()
    tag, fit2, niters2 = q.get()
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert tag == "fit"
    assert niters2 == k1.niters
    assert abs(fit2 - k1.fit) < 1e-8, (fit2, k1.fit)


def test_best_grid():
    from splatt_amd.parallel.grid import best_grid, prime_factors
    assert prime_factors(12) == [3, 2, 2]
    assert best_grid([100, 10, 10], 4) == [4, 1, 1]
    g = best_grid([40, 30, 60], 4)
    assert sorted(g) == [1, 2, 2] and g[2] == 2
    assert best_grid([10, 10, 10], 8) == [2, 2, 2]


def test_grid_localize():
    from splatt_amd.parallel.grid import GridDecomp
    t = sp.SpTensor.synthetic(DIMS, 2000, seed=9)
    total = 0
    for r in range(4):
        dec = GridDecomp.create(list(DIMS), grid=[2, 1, 2], rank=r)
        s = dec.localize(t)
        total += s.nnz
        for m in range(3):
            assert s.dims[m] == dec.chunkn[m]
            if s.nnz:
                assert int(s.inds[m].max()) < dec.chunkn[m]
    assert total == t.nnz


def _grid_worker(rank, world, file_store, result_q, out_prefix=None):
    torch.distributed.init_process_group(
        "gloo", init_method=f"file://{file_store}", rank=rank,
        world_size=world)
    try:
        from splatt_amd.parallel.grid import (GridDecomp, grid_cpd_als,
                                              write_factors)
        from splatt_amd.parallel.dist_cpd import build_shard_csf
        t = sp.SpTensor.synthetic(DIMS, NNZ, seed=SEED)
        dec = GridDecomp.create(list(DIMS), grid=[2, 1, 2])
        shard = dec.localize(t)
        cs = build_shard_csf(shard, list(DIMS), "two")
        opts = sp.CpdOptions(max_iters=ITERS, tolerance=0.0, seed=SEED)
        k = grid_cpd_als(cs, dec, RANK_F, opts)
        if out_prefix:
            write_factors(k, dec, prefix=out_prefix)
        if rank == 0:
            result_q.put(("fit", k.fit, k.niters))
    finally:
        torch.distributed.destroy_process_group()


@pytest.mark.timeout(300)
def test_grid_cpd_matches_single_process(tmp_path):
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    t = sp.SpTensor.synthetic(DIMS, NNZ, seed=SEED)
    opts = sp.CpdOptions(max_iters=ITERS, tolerance=0.0, seed=SEED)
    k1 = sp.cpd_als(t, RANK_F, opts)

    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    store = str(tmp_path / "store_grid")
    prefix = str(tmp_path / "gout_")
    procs = [ctx.Process(target=_grid_worker, args=(r, 4, store, q, prefix))
             for r in range(4)]
    for p in procs:
        p.start()
    tag, fit4, niters4 = q.get()
    for p in procs:
        p.join(timeout=180)
        assert p.exitcode == 0
    assert niters4 == k1.niters
    assert abs(fit4 - k1.fit) < 1e-8, (fit4, k1.fit)
    # rank-0 wrote GLOBAL-row factor files (chunked modes gathered)
    for m, d in enumerate(DIMS):
        lines = open(f"{prefix}mode{m + 1}.mat").read().strip().splitlines()
        assert len(lines) == d, (m, len(lines))


def _fine_worker(rank, world, file_store, result_q):
    torch.distributed.init_process_group(
        "gloo", init_method=f"file://{file_store}", rank=rank,
        world_size=world)
    try:
        from splatt_amd.parallel.grid import GridDecomp, grid_cpd_als
        from splatt_amd.parallel.dist_cpd import build_shard_csf
        t = sp.SpTensor.synthetic(DIMS, NNZ, seed=SEED)
        # arbitrary (hash-based) nnz partition, reference fine-grained style
        part = (t.inds.sum(0) * 2654435761 % 2) % world
        dec = GridDecomp.create_fine(list(DIMS), nnz_part=part)
        shard = dec.localize(t)
        cs = build_shard_csf(shard, list(DIMS), "two")
        opts = sp.CpdOptions(max_iters=ITERS, tolerance=0.0, seed=SEED)
        k = grid_cpd_als(cs, dec, RANK_F, opts)
        if rank == 0:
            result_q.put(("fit", k.fit, k.niters))
    finally:
        torch.distributed.destroy_process_group()


@pytest.mark.timeout(300)
def test_fine_cpd_matches_single_process(tmp_path):
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    t = sp.SpTensor.synthetic(DIMS, NNZ, seed=SEED)
    opts = sp.CpdOptions(max_iters=ITERS, tolerance=0.0, seed=SEED)
    k1 = sp.cpd_als(t, RANK_F, opts)

    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    store = str(tmp_path / "store_fine")
    procs = [ctx.Process(target=_fine_worker, args=(r, 2, store, q))
             for r in range(2)]
    for p in procs:
        p.start()
    tag, fit2, niters2 = q.get()
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    assert niters2 == k1.niters
    assert abs(fit2 - k1.fit) < 1e-8, (fit2, k1.fit)


def test_load_shard_and_write_factors(tmp_path):
    """Single-process path of the distributed IO helpers."""
    from splatt_amd.parallel.grid import (GridDecomp, grid_cpd_als,
                                          load_shard, write_factors)
    from splatt_amd.parallel.dist_cpd import build_shard_csf
    t = sp.SpTensor.synthetic(DIMS, 1000, seed=4)
    p = tmp_path / "t.tns"
    t.save(p)
    dec = GridDecomp.create(list(DIMS))
    shard = load_shard(str(p), dec)
    assert shard.nnz == t.nnz
    cs = build_shard_csf(shard, list(DIMS), "two")
    k = grid_cpd_als(cs, dec, 4, sp.CpdOptions(max_iters=2, tolerance=0.0))
    import os
    os.chdir(tmp_path)
    write_factors(k, dec)
    for m in range(3):
        assert (tmp_path / f"mode{m + 1}.mat").exists()
    assert (tmp_path / "lambda.mat").exists()


def _w8_worker(rank, world, file_store, result_q, grid, force_prims=False):
    if force_prims:
        os.environ["SPLATT_FORCE_RS_PRIMS"] = "1"
    torch.distributed.init_process_group(
        "gloo", init_method=f"file://{file_store}", rank=rank,
        world_size=world)
    try:
        from splatt_amd.parallel.grid import GridDecomp, grid_cpd_als
        from splatt_amd.parallel.dist_cpd import build_shard_csf
        t = sp.SpTensor.synthetic(DIMS, NNZ, seed=SEED)
        dec = GridDecomp.create(list(DIMS), grid=grid)
        shard = dec.localize(t)
        cs = build_shard_csf(shard, list(DIMS), "two")
        opts = sp.CpdOptions(max_iters=3, tolerance=0.0, seed=SEED)
        k = grid_cpd_als(cs, dec, RANK_F, opts)
        if rank == 0:
            result_q.put(k.fit)
    finally:
        torch.distributed.destroy_process_group()


@pytest.mark.timeout(600)
@pytest.mark.parametrize("grid,force_prims",
                         [([1, 1, 8], False), ([2, 1, 4], False),
                          ([1, 1, 8], True)])
def test_world8_matches_single(tmp_path, grid, force_prims):
    """The round-end 8-rank topologies (coarse + medium) over gloo; the
    coarse one also with the true RCCL reduce-scatter/all-gather
    primitives forced (SPLATT_FORCE_RS_PRIMS) — the exact SCALE_r
    configuration of the driver."""
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    t = sp.SpTensor.synthetic(DIMS, NNZ, seed=SEED)
    k1 = sp.cpd_als(t, RANK_F, sp.CpdOptions(max_iters=3, tolerance=0.0,
                                             seed=SEED))
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    store = str(tmp_path / f"store8_{grid[0]}_{grid[2]}_{force_prims}")
    procs = [ctx.Process(target=_w8_worker,
                         args=(r, 8, store, q, grid, force_prims))
             for r in range(8)]
    for p in procs:
        p.start()
    fit8 = q.get()
    for p in procs:
        p.join(timeout=300)
        assert p.exitcode == 0
    assert abs(fit8 - k1.fit) < 1e-8, (fit8, k1.fit, grid, force_prims)


def _rsag_worker(rank, world, file_store, result_q):
    # SPLATT_FORCE_RS_PRIMS: take the RCCL branch of _reduce_scatter_rows /
    # _all_gather_rows (true reduce_scatter_tensor + all_gather_into_tensor,
    # equal padded chunks) over gloo — the exact code the 8-GPU run executes.
    os.environ["SPLATT_FORCE_RS_PRIMS"] = "1"
    torch.distributed.init_process_group(
        "gloo", init_method=f"file://{file_store}", rank=rank,
        world_size=world)
    try:
        from splatt_amd.parallel.grid import (
            GridDecomp, grid_cpd_als, _reduce_scatter_rows, _all_gather_rows)
        from splatt_amd.parallel.dist_cpd import build_shard_csf

        # primitive-level check, uneven rows (n % gsize != 0 -> padding)
        for n in (10, 9, 7):
            base = torch.arange(n * 4, dtype=torch.float64).reshape(n, 4)
            full = base * (rank + 1)
            ref = base * sum(r + 1 for r in range(world))  # all-reduce result
            per = (n + world - 1) // world
            lo = min(rank * per, n)
            hi = min(lo + per, n)
            own, work = _reduce_scatter_rows(full.clone(), lo, hi,
                                             torch.distributed.group.WORLD,
                                             world)
            if work is not None:
                work.wait()
            assert torch.allclose(own, ref[lo:hi]), (n, rank)
            out = torch.empty(n, 4, dtype=torch.float64)
            _all_gather_rows(own, n, torch.distributed.group.WORLD, world, out)
            assert torch.allclose(out, ref), (n, rank)

        # end-to-end: grid CPD fit must equal the single-process fit.
        # world 4, grid [2,1,2]: mode 1 has chunkn=30, layer size 4 ->
        # per=8 with a 6-row last owner + 2 pad rows. world 3 (coarse
        # [1,1,3]): 3 divides no mode dim -> every mode remainder path.
        t = sp.SpTensor.synthetic(DIMS, NNZ, seed=SEED)
        dec = GridDecomp.create(
            list(DIMS), grid=[2, 1, 2] if world == 4 else [1, 1, 3])
        shard = dec.localize(t)
        cs = build_shard_csf(shard, list(DIMS), "two")
        opts = sp.CpdOptions(max_iters=ITERS, tolerance=0.0, seed=SEED)
        k = grid_cpd_als(cs, dec, RANK_F, opts)
        if rank == 0:
            result_q.put(("fit", k.fit))
    finally:
        torch.distributed.destroy_process_group()


@pytest.mark.timeout(300)
@pytest.mark.parametrize("world", [3, 4])
def test_grid_cpd_true_rs_primitives(tmp_path, world):
    """Force the RCCL reduce-scatter/all-gather tensor primitives (padded
    equal chunks) under gloo and require the world-N fit to equal the
    single-process fit — covers the branch only nccl takes in production.
    world=3 is a non-divisor of every mode dim (the reference tests np=7
    for the same remainder-path reason)."""
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    t = sp.SpTensor.synthetic(DIMS, NNZ, seed=SEED)
    opts = sp.CpdOptions(max_iters=ITERS, tolerance=0.0, seed=SEED)
    k1 = sp.cpd_als(t, RANK_F, opts)

    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    store = str(tmp_path / f"store_rsag{world}")
    procs = [ctx.Process(target=_rsag_worker, args=(r, world, store, q))
             for r in range(world)]
    for p in procs:
        p.start()
    tag, fit4 = q.get()
    for p in procs:
        p.join(timeout=180)
        assert p.exitcode == 0
    assert tag == "fit"
    assert abs(fit4 - k1.fit) < 1e-8, (fit4, k1.fit)


def _chunked_worker(rank, world, file_store, result_q, chunks, rsag):
    os.environ["SPLATT_COMM_CHUNKS"] = str(chunks)
    os.environ["SPLATT_COMM_CHUNK_MIN_MB"] = "0"
    if rsag:
        os.environ["SPLATT_FORCE_RS_PRIMS"] = "1"
    else:
        os.environ["SPLATT_NO_RSAG"] = "1"
    torch.distributed.init_process_group(
        "gloo", init_method=f"file://{file_store}", rank=rank,
        world_size=world)
    try:
        from splatt_amd.mttkrp import mttkrp_rows_ok
        from splatt_amd.parallel.grid import GridDecomp, grid_cpd_als
        t = sp.SpTensor.synthetic(DIMS, NNZ, seed=SEED)
        dec = GridDecomp.create(list(DIMS), grid=[2, 1, 2])
        shard = dec.localize(t)
        cs = build_shard_csf(shard, list(DIMS), "all")  # all depths root
        # the chunked pipeline must actually engage on every mode
        assert all(mttkrp_rows_ok(cs, m, RANK_F) for m in range(3))
        opts = sp.CpdOptions(max_iters=ITERS, tolerance=0.0, seed=SEED)
        k = grid_cpd_als(cs, dec, RANK_F, opts)
        if rank == 0:
            result_q.put(("fit", k.fit))
    finally:
        torch.distributed.destroy_process_group()


@pytest.mark.timeout(300)
@pytest.mark.parametrize("rsag", [True, False])
def test_chunked_comm_pipeline_fit_exact(tmp_path, rsag):
    """The chunked MTTKRP -> reduce-scatter pipeline (SPLATT_COMM_CHUNKS=3,
    overlapping each chunk's collective under the next chunk's compute)
    produces the same fit as the unchunked schedule (VERDICT r1 item 2:
    overlap must be fit-exact), in both the RS/AG and all-reduce comms."""
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    fits = {}
    for chunks in (1, 3):
        ctx = mp.get_context("spawn")
        q = ctx.SimpleQueue()
        store = str(tmp_path / f"store_ck{chunks}_{rsag}")
        procs = [ctx.Process(target=_chunked_worker,
                             args=(r, 4, store, q, chunks, rsag))
                 for r in range(4)]
        for p in procs:
            p.start()
        tag, fit = q.get()
        for p in procs:
            p.join(timeout=180)
            assert p.exitcode == 0
        fits[chunks] = fit
    assert abs(fits[1] - fits[3]) < 1e-10, fits


@pytest.mark.timeout(420)
def test_dist_fuzz_bounded():
    """Bounded randomized distributed fuzz (4 trials of
    scripts/dist_fuzz.py) in CI — random worlds/grids/schedules, fit
    equal to single-process."""
    import subprocess
    import sys
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ, MASTER_ADDR="127.0.0.1",
               PYTHONPATH=root + os.pathsep + os.environ.get("PYTHONPATH", ""))
    r = subprocess.run([sys.executable, "scripts/dist_fuzz.py", "4", "99"],
                       capture_output=True, text=True, cwd=root, env=env,
                       timeout=400)
    assert r.returncode == 0, (r.stdout[-500:], r.stderr[-800:])
    assert "dist fuzz clean" in r.stdout
