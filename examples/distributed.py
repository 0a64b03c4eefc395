"""Distributed CPD over torch.distributed (RCCL on GPUs, gloo on CPU).

    python -m torch.distributed.run --standalone --nnodes=1 \
        --nproc-per-node 4 --local-addr 127.0.0.1 examples/distributed.py
"""
import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))
import os

import torch
import torch.distributed as dist

import splatt_amd as sp
from splatt_amd.parallel.dist_cpd import build_shard_csf
from splatt_amd.parallel.grid import (GridDecomp, comm_stats, grid_cpd_als,
                                      write_factors)


def main():
    use_cuda = torch.cuda.is_available()
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if use_cuda:
        torch.cuda.set_device(local_rank % torch.cuda.device_count())
    dist.init_process_group("nccl" if use_cuda else "gloo")
    rank = dist.get_rank()

    # every rank sees the same global tensor (same seed), keeps its box
    dims = [400, 300, 600]
    t = sp.SpTensor.synthetic(dims, 100_000, seed=99)
    dec = GridDecomp.create(dims)           # auto nmodes-D grid
    shard = dec.localize(t)
    if use_cuda:
        shard = shard.to("cuda")
    cs = build_shard_csf(shard, dims, "all", flat_only=use_cuda)

    stats = comm_stats(dec, shard.nnz, 16)   # collective: call on ALL ranks
    if rank == 0:
        print("grid:", dec.grid, "| comm:", stats)
    k = grid_cpd_als(cs, dec, 16, sp.CpdOptions(max_iters=10))
    if rank == 0:
        print(f"fit = {k.fit:.5f} (identical to a single-process run)")
    write_factors(k, dec, prefix="/tmp/example_")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
