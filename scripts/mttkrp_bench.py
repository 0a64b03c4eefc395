#!/usr/bin/env python3
"""BASELINE config 2: pure MTTKRP kernel throughput on the NELL-2 shape
(per-mode, production dispatch + comparison algorithms)."""
import time

import torch

import splatt_amd as sp
from splatt_amd.mttkrp import mttkrp
from splatt_amd.parallel.dist_cpd import build_shard_csf

DIMS = [12092, 9184, 28818]
NNZ = 76_879_419
RANK = 16
REPS = 30


def timeit(fn):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    tic = time.time()
    for _ in range(REPS):
        fn()
    torch.cuda.synchronize()
    return (time.time() - tic) / REPS


def main():
    t = sp.SpTensor.synthetic(DIMS, NNZ, seed=0xB0B0)
    td = t.to("cuda")
    mats = [sp.seeded_init(d, RANK, m, 123).cuda()
            for m, d in enumerate(DIMS)]
    flops = 3.0 * NNZ * RANK

    cs = build_shard_csf(td, DIMS, "all", flat_only=True, stage_rank=RANK)
    for mode in range(3):
        out = torch.empty(DIMS[mode], RANK, dtype=torch.float64,
                          device="cuda")
        ms = timeit(lambda: mttkrp(cs, mats, mode, out=out)) * 1e3
        print(f"flat+LDS v6 (production) mode {mode}: {ms:8.3f} ms "
              f"{flops / ms / 1e6:8.1f} GFLOP/s", flush=True)
    for mode in range(3):
        out = torch.empty(DIMS[mode], RANK, dtype=torch.float64,
                          device="cuda")
        ms = timeit(lambda: mttkrp(cs, mats, mode, out=out,
                                   deterministic=True)) * 1e3
        print(f"det6 deterministic    mode {mode}: {ms:8.3f} ms "
              f"{flops / ms / 1e6:8.1f} GFLOP/s", flush=True)
    cs2 = build_shard_csf(td, DIMS, "all", flat_only=True)
    for mode in range(3):
        out = torch.empty(DIMS[mode], RANK, dtype=torch.float64,
                          device="cuda")
        ms = timeit(lambda: mttkrp(cs2, mats, mode, out=out)) * 1e3
        print(f"flat v2 (no staging)  mode {mode}: {ms:8.3f} ms "
              f"{flops / ms / 1e6:8.1f} GFLOP/s", flush=True)


if __name__ == "__main__":
    main()
