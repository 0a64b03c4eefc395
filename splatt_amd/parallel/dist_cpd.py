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

import math
from dataclasses import dataclass
from typing import List, Optional

import torch
import torch.distributed as dist

from splatt_amd.cpd import CpdOptions, Kruskal, seeded_init, _normalize
from splatt_amd.csf import CsfSet, build_csf, order_modes
from splatt_amd.mttkrp import mttkrp
from splatt_amd.ops.dense import gram
from splatt_amd.sptensor import SpTensor


def _world() -> int:
    return dist.get_world_size() if dist.is_available() and dist.is_initialized() else 1


def _all_reduce(t: torch.Tensor, op=None) -> None:
    if _world() > 1:
        dist.all_reduce(t, op=op or dist.ReduceOp.SUM)


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


@dataclass
class DistCpdState:
    """Device-resident distributed ALS state, reusable across timed steps."""
    cs: CsfSet
    part_mode: int
    row0: int
    global_dims: List[int]
    factors: List[torch.Tensor]
    grams: List[torch.Tensor]
    lam: torch.Tensor
    buf: torch.Tensor
    norm_x: float
    fit: float = 0.0
    old_fit: float = 0.0
    niters: int = 0


def dist_cpd_init(shard_cs: CsfSet, part_mode: int, row0: int,
                  global_dims: List[int], rank_f: int,
                  opts: CpdOptions) -> DistCpdState:
    nm = len(global_dims)
    dev = shard_cs.csfs[0].device
    dtype = shard_cs.csfs[0].vals.dtype
    local_dims = shard_cs.dims

    factors = []
    for m in range(nm):
        if m == part_mode:
            f = seeded_init(local_dims[m], rank_f, m, opts.seed, row0=row0,
                            dtype=dtype)
        else:
            f = seeded_init(global_dims[m], rank_f, m, opts.seed, dtype=dtype)
        factors.append(f.to(dev))

    grams = []
    for m in range(nm):
        g = gram(factors[m])
        if m == part_mode:
            _all_reduce(g)
        grams.append(g)

    nx = torch.tensor([float(shard_cs.csfs[0].vals.double().square().sum())],
                      dtype=torch.float64, device=dev)
    _all_reduce(nx)

    maxdim = max(max(global_dims), local_dims[part_mode])
    return DistCpdState(
        cs=shard_cs, part_mode=part_mode, row0=row0,
        global_dims=list(global_dims), factors=factors,
        grams=grams, lam=torch.ones(factors[0].shape[1], dtype=dtype, device=dev),
        buf=torch.empty(maxdim, factors[0].shape[1], dtype=dtype, device=dev),
        norm_x=float(nx.item()))


def dist_cpd_step(st: DistCpdState, it: int) -> float:
    """One full ALS iteration (all modes + fit). Returns the fit."""
    nm = len(st.global_dims)
    q = st.part_mode
    dev = st.buf.device
    dtype = st.buf.dtype
    F = st.factors[0].shape[1]
    local_dims = st.cs.dims
    eye = torch.eye(F, dtype=dtype, device=dev)

    for m in range(nm):
        nrows = local_dims[m] if m == q else st.global_dims[m]
        mb = st.buf[:nrows]
        mttkrp(st.cs, st.factors, m, out=mb)
        if m != q:
            # sum partial contributions from every layer (RCCL over xGMI)
            _all_reduce(mb)
        G = torch.ones(F, F, dtype=dtype, device=dev)
        for o in range(nm):
            if o != m:
                G *= st.grams[o]
        L = torch.linalg.cholesky(G + 1e-12 * G.diagonal().abs().max() * eye)
        A = mb @ torch.cholesky_inverse(L)
        # normalize with GLOBAL column norms for the partitioned mode
        if m == q:
            if it == 0:
                sq = A.square().sum(dim=0)
                _all_reduce(sq)
                lam = sq.sqrt()
            else:
                mx = A.abs().amax(dim=0)
                _all_reduce(mx, op=dist.ReduceOp.MAX)
                lam = mx.clamp_(min=1.0)
            lam = torch.where(lam == 0, torch.ones_like(lam), lam)
            A /= lam
        else:
            lam = _normalize(A, it)
        st.lam = lam
        st.factors[m] = A
        g = gram(A)
        if m == q:
            _all_reduce(g)
        st.grams[m] = g

    # fit (reference mpi fit: local inner + 1-double allreduce, mpi_cpd.c:94)
    mlast = nm - 1
    nrows = local_dims[mlast] if mlast == q else st.global_dims[mlast]
    inner_t = (st.buf[:nrows].double() * st.factors[mlast].double()).sum(dim=0) \
        @ st.lam.double()
    if mlast == q:
        _all_reduce(inner_t)
    inner = float(inner_t)
    Gall = torch.ones(F, F, dtype=dtype, device=dev)
    for o in range(nm):
        Gall *= st.grams[o]
    knorm = float(st.lam.double() @ Gall.double() @ st.lam.double())
    residual = math.sqrt(max(0.0, st.norm_x + knorm - 2 * inner))
    st.old_fit = st.fit
    st.fit = 1.0 - residual / math.sqrt(st.norm_x)
    st.niters = it + 1
    return st.fit


def dist_cpd_als(shard_cs: CsfSet, part_mode: int, row0: int,
                 global_dims: List[int], rank_f: int,
                 opts: Optional[CpdOptions] = None) -> Kruskal:
    """Full distributed CPD-ALS loop with convergence check."""
    opts = opts or CpdOptions()
    st = dist_cpd_init(shard_cs, part_mode, row0, global_dims, rank_f, opts)
    trace = []
    for it in range(opts.max_iters):
        fit = dist_cpd_step(st, it)
        trace.append(fit)
        if opts.verbose and (_world() == 1 or dist.get_rank() == 0):
            print(f"  its = {it + 1} fit = {fit:.5f}", flush=True)
        if it > 0 and abs(fit - st.old_fit) < opts.tolerance:
            break
    return Kruskal(factors=st.factors, lam=st.lam, fit=st.fit,
                   niters=st.niters, fit_trace=trace)
