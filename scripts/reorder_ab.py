#!/usr/bin/env python3
"""A/B: does BFS / degree-sort reordering speed up the GPU MTTKRP on a
STRUCTURED (zipf) tensor? (evaluate reordering on
structured shapes, not uniform synthetics; reference reorder.c:412-462.)
Prints per-variant full-ALS-step times on a NELL-shaped zipf tensor."""
import sys
import time

import torch

import splatt_amd as sp
from splatt_amd.parallel.dist_cpd import build_shard_csf
from splatt_amd.parallel.grid import GridDecomp, grid_cpd_init, grid_cpd_step
from splatt_amd.reorder import perm_apply, perm_bfs, Permutation

DIMS = [12092, 9184, 28818]
NNZ = 40_000_000
RANK = 16
STEPS = 10


def bench(t, tag):
    dec = GridDecomp.create(list(t.dims))
    cs = build_shard_csf(t.to("cuda"), list(t.dims), "all",
                         flat_only=True, stage_rank=RANK)
    st = grid_cpd_init(cs, dec, RANK, sp.CpdOptions(max_iters=STEPS + 3,
                                                    tolerance=0.0))
    it = 0
    for _ in range(3):
        grid_cpd_step(st, it)
        it += 1
    torch.cuda.synchronize()
    tic = time.time()
    for _ in range(STEPS):
        grid_cpd_step(st, it)
        it += 1
    torch.cuda.synchronize()
    ms = (time.time() - tic) / STEPS * 1e3
    print(f"{tag:24s} {ms:8.3f} ms/step", flush=True)
    return ms


def degree_sort_perm(t):
    """Relabel each mode by descending nnz count (hot rows first -> hot
    factor rows share cache lines)."""
    perms, iperms = [], []
    for m in range(t.nmodes):
        cnt = torch.bincount(t.inds[m], minlength=t.dims[m])
        order = torch.argsort(cnt, descending=True, stable=True)
        iperm = torch.empty_like(order)
        iperm[order] = torch.arange(t.dims[m])
        perms.append(order)
        iperms.append(iperm)
    return Permutation(perms=perms, iperms=iperms)


def main():
    t = sp.SpTensor.synthetic(DIMS, NNZ, seed=7, dist="zipf").fixed(
        dedup=True)
    print(f"# zipf tensor nnz={t.nnz}", flush=True)
    base = bench(t, "baseline")
    tb = perm_apply(t, perm_bfs(t))
    bfs = bench(tb, "bfs-reordered")
    td = perm_apply(t, degree_sort_perm(t))
    deg = bench(td, "degree-sorted")
    print(f"# speedup bfs {base / bfs:.3f}x  degree {base / deg:.3f}x",
          flush=True)


if __name__ == "__main__":
    sys.exit(main())
