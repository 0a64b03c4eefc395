#!/usr/bin/env python3
"""Randomized GPU MTTKRP/CPD fuzz: random shapes x ranks x build flavors
x dispatch modes, every result checked against the CPU oracle. Exits
nonzero on first mismatch (prints the repro config)."""
import random
import sys

import torch

import splatt_amd as sp
from splatt_amd.mttkrp import mttkrp, mttkrp_rows_ok
from splatt_amd.parallel.dist_cpd import build_shard_csf

SEED = int(sys.argv[1]) if len(sys.argv) > 1 else 1234
ROUNDS = int(sys.argv[2]) if len(sys.argv) > 2 else 40
rng = random.Random(SEED)


def one(i):
    nm = rng.choice([3, 3, 4, 5])
    dims = [rng.randint(2, 900) for _ in range(nm)]
    nnz = rng.randint(50, 60_000)
    rank = rng.choice([4, 8, 16, 32, 64, 7, 10, 24])
    policy = rng.choice(["one", "two", "all"])
    flat = rng.choice([True, False])
    stage = rng.choice([0, rank]) if flat else 0
    cfg = dict(i=i, dims=dims, nnz=nnz, rank=rank, policy=policy,
               flat=flat, stage=stage)
    dtype = rng.choice([torch.float64, torch.float64, torch.float32])
    cfg["dtype"] = str(dtype)
    t = sp.SpTensor.synthetic(dims, nnz, seed=SEED + i,
                              dtype=dtype).fixed(dedup=True)
    mats_c = [sp.seeded_init(d, rank, m, 5 + i, dtype=dtype)
              for m, d in enumerate(dims)]
    mats_g = [m.cuda() for m in mats_c]
    cs = build_shard_csf(t.to("cuda"), dims, policy, flat_only=flat,
                         gather_tiles=0, stage_rank=stage)
    tol = 1e-8 if dtype == torch.float64 else 1e-2
    for mode in range(nm):
        ref = sp.mttkrp_stream(t, mats_c, mode)
        out = mttkrp(cs, mats_g, mode)
        scale = max(1.0, float(ref.abs().max()))
        err = (out.cpu() - ref).abs().max().item()
        assert err < tol * scale, (cfg, mode, "default", err)
        # rows-restricted tiling when supported
        if mttkrp_rows_ok(cs, mode, rank):
            o2 = torch.empty_like(out)
            n = dims[mode]
            cuts = sorted({0, n, rng.randint(0, n), rng.randint(0, n)})
            for a, b in zip(cuts, cuts[1:]):
                mttkrp(cs, mats_g, mode, out=o2, rows=(a, b))
            err = (o2 - out).abs().max().item()
            # split-vs-fused summation order differs; fp-level agreement
            rtol = 1e-9 if dtype == torch.float64 else 1e-3
            assert err < rtol * scale, (cfg, mode, "rows", err)
        # reduced-precision factor storage where legal (spec rank, f64)
        if (dtype == torch.float64 and rank in (4, 8, 16, 32, 64)
            and rng.random() < 0.3):
            import os as _os
            store = rng.choice(["f32", "bf16"])
            qdt = {"f32": torch.float32,
                   "bf16": torch.bfloat16}[store]
            _os.environ["SPLATT_FACTOR_STORE"] = store
            try:
                oq = mttkrp(cs, [m.to(qdt) for m in mats_g], mode)
            finally:
                del _os.environ["SPLATT_FACTOR_STORE"]
            stol = 1e-4 if store == "f32" else 5e-2
            err = (oq.cpu() - ref).abs().max().item()
            assert err < stol * scale, (cfg, mode, store, err)
        # deterministic path where legal (depth-0 + spec rank + <=5 modes)
        depth = cs.mode_depth[mode]
        if depth == 0 and rank in (4, 8, 16, 32, 64) and nm <= 5:
            d1 = mttkrp(cs, mats_g, mode, deterministic=True)
            d2 = mttkrp(cs, mats_g, mode, deterministic=True)
            assert torch.equal(d1, d2), (cfg, mode, "det-repeat")
            err = (d1.cpu() - ref).abs().max().item()
            assert err < tol * scale, (cfg, mode, "det", err)
    return cfg


for i in range(ROUNDS):
    cfg = one(i)
    if i % 10 == 0:
        print("ok", cfg, flush=True)
print(f"fuzz clean: {ROUNDS} configs, seed {SEED}")
