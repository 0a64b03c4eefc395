#!/usr/bin/env python3
"""Randomized distributed-layer fuzz (gloo, CPU): random shapes, worlds,
grids, comm schedules and pipeline depths; every trial's world-N fit must
match the single-process fit (the rank-invariance property). Usage:
dist_fuzz.py [trials] [seed]."""
import os
import random
import sys
import tempfile

import torch
import torch.multiprocessing as mp

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import splatt_amd as sp  # noqa: E402


def worker(rank, world, store, q, cfg):
    os.environ["SPLATT_COMM_CHUNKS"] = str(cfg["chunks"])
    os.environ["SPLATT_COMM_CHUNK_MIN_MB"] = "0"
    if cfg["force_prims"]:
        os.environ["SPLATT_FORCE_RS_PRIMS"] = "1"
    if cfg["no_rsag"]:
        os.environ["SPLATT_NO_RSAG"] = "1"
    torch.distributed.init_process_group(
        "gloo", init_method=f"file://{store}", rank=rank, world_size=world)
    try:
        from splatt_amd.parallel.dist_cpd import build_shard_csf
        from splatt_amd.parallel.grid import GridDecomp, grid_cpd_als
        t = sp.SpTensor.synthetic(cfg["dims"], cfg["nnz"], seed=cfg["seed"])
        dec = GridDecomp.create(list(cfg["dims"]), grid=cfg["grid"])
        cs = build_shard_csf(dec.localize(t), list(cfg["dims"]),
                             cfg["policy"])
        k = grid_cpd_als(cs, dec, cfg["rank"],
                         sp.CpdOptions(max_iters=cfg["iters"], tolerance=0.0,
                                       seed=cfg["seed"]))
        if rank == 0:
            q.put(k.fit)
    finally:
        torch.distributed.destroy_process_group()


def factor_grid(world, nm, rng):
    grid = [1] * nm
    w = world
    while w > 1:
        for p in (2, 3, 5, 7):
            if w % p == 0:
                grid[rng.randrange(nm)] *= p
                w //= p
                break
    return grid


def main():
    trials = int(sys.argv[1]) if len(sys.argv) > 1 else 12
    seed = int(sys.argv[2]) if len(sys.argv) > 2 else 5
    rng = random.Random(seed)
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    for i in range(trials):
        nm = rng.choice([3, 3, 4])
        cfg = dict(
            dims=[rng.randint(6, 60) for _ in range(nm)],
            nnz=rng.randint(200, 6000),
            rank=rng.choice([4, 6, 8]),
            iters=rng.randint(2, 5),
            seed=seed * 1000 + i,
            world=rng.choice([2, 3, 4]),
            policy=rng.choice(["one", "two", "all"]),
            chunks=rng.choice([1, 2, 3]),
            force_prims=rng.random() < 0.5,
            no_rsag=rng.random() < 0.3,
        )
        cfg["grid"] = rng.choice(
            [None, factor_grid(cfg["world"], nm, rng)])
        t = sp.SpTensor.synthetic(cfg["dims"], cfg["nnz"], seed=cfg["seed"])
        ref = sp.cpd_als(t, cfg["rank"],
                         sp.CpdOptions(max_iters=cfg["iters"], tolerance=0.0,
                                       seed=cfg["seed"]))
        ctx = mp.get_context("spawn")
        q = ctx.SimpleQueue()
        store = tempfile.mktemp(prefix="dfz")
        procs = [ctx.Process(target=worker,
                             args=(r, cfg["world"], store, q, cfg))
                 for r in range(cfg["world"])]
        for p in procs:
            p.start()
        fit = q.get()
        for p in procs:
            p.join(timeout=120)
            assert p.exitcode == 0, (cfg, p.exitcode)
        err = abs(fit - ref.fit)
        assert err < 1e-8, (cfg, fit, ref.fit)
        print(f"ok {i}: world={cfg['world']} grid={cfg['grid']} "
              f"chunks={cfg['chunks']} prims={cfg['force_prims']} "
              f"rsag={not cfg['no_rsag']} err={err:.2e}", flush=True)
    print(f"dist fuzz clean: {trials} trials, seed {seed}")


if __name__ == "__main__":
    main()
