"""Distributed CPD-ALS over torch.distributed (RCCL on ROCm, gloo on CPU).

Capability parity: the reference's distributed CPD layer
(src/mpi/mpi_cpd.c:627-804 mpi_cpd_als_iterate; row exchange
mpi_update_rows/mpi_reduce_rows mpi_cpd.c:807-890; Gram/lambda allreduce
matrix.c:448/121). Fresh MI355X design — one process per GPU on one node,
xGMI collectives instead of MPI point-to-point:

  * the tensor is layer-partitioned along one mode q (the reference's
    coarse/medium layer boundaries, mpi_io.c:365-439): rank r owns the
    contiguous slice block [row0, row0+nlocal) of mode q and ALL nonzeros
    whose q-index falls there — so mode-q MTTKRP output rows are owned
    exclusively and need no communication at all.
  * for every other mode the factor is replicated; partial MTTKRP outputs
    are summed with one bucketed all_reduce (RCCL ring over xGMI) — the
    semantic fusion of the reference's alltoallv reduce+update pair when
    every rank needs every row (SURVEY.md §2.4 mapping).
  * Gram matrices (F x F) and lambda (F) ride tiny all_reduces.
  * factor init is partition-invariant (seeded_factor_init), so fit at
    world N equals fit at world 1 for the same seed — the reference's
    mpi_mat_rand rank-invariance property, made testable.
"""
from __future__ import annotations

from typing import List, Optional

import torch

from splatt_amd.cpd import CpdOptions, Kruskal
from splatt_amd.csf import CsfSet, build_csf, order_modes
from splatt_amd.sptensor import SpTensor


def partition_rows(n: int, world: int, rank: int) -> tuple[int, int]:
    """Contiguous near-equal row blocks (remainder to the low ranks)."""
    base, rem = divmod(n, world)
    row0 = rank * base + min(rank, rem)
    nloc = base + (1 if rank < rem else 0)
    return row0, nloc


def localize_shard(t: SpTensor, part_mode: int, row0: int, nloc: int) -> SpTensor:
    """Select nonzeros whose part_mode index is in [row0, row0+nloc) and
    shift that mode to local coordinates (reference index localization,
    mpi_io.c:756-844)."""
    mask = (t.inds[part_mode] >= row0) & (t.inds[part_mode] < row0 + nloc)
    inds = t.inds[:, mask].clone()
    inds[part_mode] -= row0
    dims = list(t.dims)
    dims[part_mode] = nloc
    return SpTensor(inds, t.vals[mask].clone(), dims)


def build_shard_csf(shard: SpTensor, global_dims: List[int],
                    policy: str = "two", flat_only: bool = False,
                    gather_tiles: int = 0, stage_rank: int = 0) -> CsfSet:
    """CSF for a shard with mode ORDER decided by the global dims, so every
    rank picks the same kernel dispatch (root/intl/leaf) per mode."""
    nm = shard.nmodes
    if policy == "one":
        perm = order_modes(global_dims, "smallfirst")
        c = build_csf(shard, perm, flat_only, gather_tiles, stage_rank)
        return CsfSet([c], [0] * nm, [c.level_of_mode(m) for m in range(nm)])
    if policy == "two":
        perm = order_modes(global_dims, "smallfirst")
        longest = perm[-1]
        c0 = build_csf(shard, perm, flat_only, gather_tiles, stage_rank)
        c1 = build_csf(shard, order_modes(global_dims, "root", longest),
                       flat_only, gather_tiles, stage_rank)
        mode_csf, mode_depth = [], []
        for m in range(nm):
            if m == longest:
                mode_csf.append(1)
                mode_depth.append(0)
            else:
                mode_csf.append(0)
                mode_depth.append(c0.level_of_mode(m))
        return CsfSet([c0, c1], mode_csf, mode_depth)
    csfs = [build_csf(shard, order_modes(global_dims, "root", m), flat_only,
                      gather_tiles, stage_rank)
            for m in range(nm)]
    return CsfSet(csfs, list(range(nm)), [0] * nm)


def _coarse_dec(part_mode: int, global_dims: List[int]):
    from splatt_amd.parallel.grid import GridDecomp, _world
    grid = [1] * len(global_dims)
    grid[part_mode] = _world()
    return GridDecomp.create(list(global_dims), grid=grid)


def dist_cpd_init(shard_cs: CsfSet, part_mode: int, row0: int,
                  global_dims: List[int], rank_f: int, opts: CpdOptions):
    """Coarse (1-D layer) decomposition = the grid driver with
    grid[part_mode] = world (kept as the historical entry point)."""
    from splatt_amd.parallel.grid import grid_cpd_init
    dec = _coarse_dec(part_mode, global_dims)
    assert dec.chunk0[part_mode] == row0, (dec.chunk0, row0)
    return grid_cpd_init(shard_cs, dec, rank_f, opts)


def dist_cpd_step(st, it: int) -> float:
    from splatt_amd.parallel.grid import grid_cpd_step
    return grid_cpd_step(st, it)


def dist_cpd_als(shard_cs: CsfSet, part_mode: int, row0: int,
                 global_dims: List[int], rank_f: int,
                 opts: Optional[CpdOptions] = None) -> Kruskal:
    from splatt_amd.parallel.grid import grid_cpd_als
    dec = _coarse_dec(part_mode, global_dims)
    assert dec.chunk0[part_mode] == row0, (dec.chunk0, row0)
    return grid_cpd_als(shard_cs, dec, rank_f, opts)
