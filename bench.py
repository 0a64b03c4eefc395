#!/usr/bin/env python3
"""Flagship benchmark: CPD-ALS over the gfx950 HIP MTTKRP on synthetic
tensors shaped like the BASELINE.json configs.

One "step" = one full ALS iteration (all modes: MTTKRP + Cholesky solve +
normalize + Gram update + fit), the reference's headline unit
(cpd_als_iterate, src/cpd.c:271). Reported value = effective MTTKRP GFLOP/s
over the WHOLE step time (3*nnz*rank flops per MTTKRP, nmodes MTTKRPs per
step — the flop convention of BASELINE.md), aggregated across all ranks.

Launch (driver contract):
  python bench.py --gpus 1 --steps K --warmup W
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

Weak scaling: the partitioned mode's dimension grows with N and every rank
generates (and owns) one full-size layer shard, so per-GPU work is fixed.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch
import torch.distributed as dist

import splatt_amd as sp
from splatt_amd.parallel.dist_cpd import (
    build_shard_csf, dist_cpd_init, dist_cpd_step, _world)

CONFIGS = {
    # name: (dims, nnz per shard, rank, concentration)
    "nell2": ([12092, 9184, 28818], 76_879_419, 16, 1.5),
    "netflix": ([480_189, 17_770, 2182], 100_480_507, 32, 1.5),
    "amazon": ([4_821_207, 1_774_269, 1_805_187], 1_741_809_018, 16, 1.5),
    "delicious4d": ([532_924, 17_262_471, 2_480_308, 1443], 140_126_181, 32, 1.5),
    "small": ([1200, 900, 1500], 300_000, 16, 1.5),
}


def synth_shard(dims, nnz, part_mode, row0, nloc, seed, dtype, device):
    """Generate this rank's layer shard directly at full local size.

    Per-GPU shard shape == the named config shape (weak scaling: the global
    tensor is N stacked layers along part_mode)."""
    local_dims = list(dims)
    local_dims[part_mode] = nloc
    gen_dev = device if nnz > 200_000_000 else "cpu"
    g = torch.Generator(device=gen_dev).manual_seed(seed)
    cols = []
    for m, d in enumerate(local_dims):
        u = torch.rand(nnz, generator=g, dtype=torch.float32, device=gen_dev)
        u = u * u.sqrt()  # mild concentration -> non-uniform fiber lengths
        cols.append((u.double() * d).long().clamp_(0, d - 1))
    inds = torch.stack(cols, 0)
    vals = torch.rand(nnz, generator=g, dtype=torch.float32,
                      device=gen_dev).to(dtype)
    return sp.SpTensor(inds, vals, local_dims).to(device)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--config", default="nell2", choices=list(CONFIGS))
    ap.add_argument("--rank-f", type=int, default=0, help="override CP rank")
    ap.add_argument("--dtype", default="f64", choices=["f64", "f32"])
    ap.add_argument("--device", default="cuda")
    ap.add_argument("--csf", default="all", choices=["one", "two", "all"])
    args = ap.parse_args()

    dims, nnz_shard, rank_f, conc = CONFIGS[args.config]
    if args.rank_f:
        rank_f = args.rank_f
    dtype = torch.float64 if args.dtype == "f64" else torch.float32

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if world > 1:
        torch.cuda.set_device(local_rank)
        dist.init_process_group("nccl")
    device = torch.device(args.device if args.device == "cpu"
                          else f"cuda:{local_rank}")

    # weak scaling: global partitioned-mode dim = N x shard dim
    part_mode = max(range(len(dims)), key=lambda m: dims[m])
    global_dims = list(dims)
    global_dims[part_mode] *= world
    nloc = dims[part_mode]
    row0 = rank * nloc

    t0 = time.time()
    shard = synth_shard(global_dims, nnz_shard, part_mode, row0, nloc,
                        seed=0xB0B0 + rank, dtype=dtype, device=device)
    cs = build_shard_csf(shard, global_dims, args.csf).to(device)
    del shard
    if rank == 0:
        print(f"# setup: shard nnz={nnz_shard} csf_bytes={cs.storage_bytes()}"
              f" build_s={time.time() - t0:.1f}", file=sys.stderr, flush=True)

    opts = sp.CpdOptions(max_iters=args.warmup + args.steps, tolerance=0.0,
                         seed=0x5EED)
    st = dist_cpd_init(cs, part_mode, row0, global_dims, rank_f, opts)

    def barrier_sync():
        if world > 1:
            dist.barrier()
        if device.type == "cuda":
            torch.cuda.synchronize(device)

    it = 0
    for _ in range(args.warmup):
        dist_cpd_step(st, it)
        it += 1
    barrier_sync()
    tic = time.time()
    for _ in range(args.steps):
        dist_cpd_step(st, it)
        it += 1
    barrier_sync()
    elapsed = time.time() - tic

    # max over ranks
    if world > 1:
        e = torch.tensor([elapsed], dtype=torch.float64, device=device)
        dist.all_reduce(e, op=dist.ReduceOp.MAX)
        elapsed = float(e.item())

    nmodes = len(global_dims)
    nnz_global = nnz_shard * world
    flops = args.steps * nmodes * 3.0 * nnz_global * rank_f
    gflops = flops / elapsed / 1e9
    if rank == 0:
        result = {
            "metric": "mttkrp_gflops",
            "value": round(gflops, 2),
            "unit": "GFLOP/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1e3, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": args.dtype,
            "data": "synthetic",
            "config": {
                "model": f"{args.config}-shaped CPD-ALS",
                "dims": global_dims,
                "nnz": nnz_global,
                "cp_rank": rank_f,
                "csf": args.csf,
                "parallelism": f"layer-partition x{world} (RCCL/xGMI)",
                "fit": round(st.fit, 6),
            },
        }
        print(json.dumps(result), flush=True)
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
