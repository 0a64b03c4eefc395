"""Medium-grained (Cartesian grid) decomposition over torch.distributed.

Capability parity: the reference's medium-grained MPI decomposition —
nmodes-D rank grid with per-mode layer communicators (mpi_setup.c:201-243),
auto grid-dim selection by prime factorization balancing mode lengths
(p_get_best_mpi_dim, mpi_io.c:537-574), nnz-box ownership + index
localization (mpi_io.c:756-844). MI355X design: RCCL groups over xGMI via
dist.new_group; the heavy mode-m partial-row reduction runs only inside
layer_m (ranks sharing the mode-m chunk), and the chunked factor means no
rank ever holds a full long-mode factor (SURVEY.md §5 long-dimension
story). The coarse/1D decomposition is the special case grid[q] = world.

Replicated-vs-chunked algebra: factor m is CHUNKED iff grid[m] > 1 (each
layer holds one chunk, replicated across the W/grid[m] layer members).
Chunk-level Gram/lambda/fit contributions are divided by the replication
factor and summed with a world all-reduce.
"""
from __future__ import annotations

import math
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from splatt_amd.cpd import CpdOptions, Kruskal, seeded_init
from splatt_amd.csf import CsfSet
from splatt_amd.mttkrp import mttkrp, mttkrp_rows_ok
from splatt_amd.ops.dense import gram, solve_rows, spd_inverse
from splatt_amd.sptensor import SpTensor


def _world() -> int:
    return dist.get_world_size() if dist.is_available() and dist.is_initialized() else 1


def prime_factors(n: int) -> List[int]:
    out = []
    d = 2
    while d * d <= n:
        while n % d == 0:
            out.append(d)
            n //= d
        d += 1
    if n > 1:
        out.append(n)
    return sorted(out, reverse=True)


def best_grid(dims: List[int], world: int) -> List[int]:
    """Assign the prime factors of `world` to the modes with the longest
    per-rank chunk (reference p_get_best_mpi_dim, mpi_io.c:537-574)."""
    grid = [1] * len(dims)
    for p in prime_factors(world):
        m = max(range(len(dims)), key=lambda i: dims[i] / grid[i])
        grid[m] *= p
    return grid


def chunk_range(dim: int, parts: int, coord: int) -> tuple[int, int]:
    base, rem = divmod(dim, parts)
    lo = coord * base + min(coord, rem)
    return lo, base + (1 if coord < rem else 0)


@dataclass
class GridDecomp:
    grid: List[int]
    rank: int
    coords: List[int]
    chunk0: List[int]
    chunkn: List[int]
    global_dims: List[int]
    layer_groups: Dict[int, Optional[object]] = field(default_factory=dict)
    layer_ranks: Dict[int, list] = field(default_factory=dict)

    @staticmethod
    def create_fine(global_dims: List[int],
                    nnz_part: Optional[torch.Tensor] = None) -> "FineDecomp":
        """Fine-grained decomposition: an arbitrary partition of the
        NONZEROS (from an external partition file, or contiguous slices);
        every factor is replicated and every mode's partial rows are
        world-all-reduced (reference fine-grained mode, mpi_setup.c:181-193,
        mpi_io.c:486-499)."""
        world = _world()
        r = dist.get_rank() if world > 1 else 0
        d = FineDecomp(grid=[1] * len(global_dims), rank=r,
                       coords=[0] * len(global_dims),
                       chunk0=[0] * len(global_dims),
                       chunkn=list(global_dims),
                       global_dims=list(global_dims),
                       nnz_part=nnz_part)
        for m in range(len(global_dims)):
            d.layer_groups[m] = None if world == 1 else dist.group.WORLD
            d.layer_ranks[m] = list(range(world))
        return d

    @staticmethod
    def create(global_dims: List[int], grid: Optional[List[int]] = None,
               rank: Optional[int] = None) -> "GridDecomp":
        world = _world()
        r = dist.get_rank() if world > 1 else 0
        if rank is not None:
            r = rank
            if world == 1 and grid is not None:
                world = math.prod(grid)  # offline decomposition planning
        grid = grid or best_grid(list(global_dims), world)
        assert math.prod(grid) == world, (grid, world)
        # row-major rank -> coords
        coords = []
        rem = r
        for g in reversed(grid):
            coords.append(rem % g)
            rem //= g
        coords.reverse()
        chunk0, chunkn = [], []
        for m, g in enumerate(grid):
            lo, n = chunk_range(global_dims[m], g, coords[m])
            chunk0.append(lo)
            chunkn.append(n)
        d = GridDecomp(grid=list(grid), rank=r, coords=coords, chunk0=chunk0,
                       chunkn=chunkn, global_dims=list(global_dims))
        d._make_groups()
        return d

    def _make_groups(self):
        """layer_m(v) = ranks whose mode-m coordinate is v: the reduction
        group for mode-m partial rows (reference layer_comm[m])."""
        world = _world()
        if world == 1:
            for m in range(len(self.grid)):
                self.layer_groups[m] = None
            return
        nm = len(self.grid)
        strides = [1] * nm
        for m in reversed(range(nm - 1)):
            strides[m] = strides[m + 1] * self.grid[m + 1]
        for m in range(nm):
            if self.grid[m] == 1:
                self.layer_groups[m] = dist.group.WORLD
                self.layer_ranks[m] = list(range(world))
                continue
            mine = None
            for v in range(self.grid[m]):
                ranks = [r for r in range(world)
                         if (r // strides[m]) % self.grid[m] == v]
                g = dist.new_group(ranks) if len(ranks) > 1 else "solo"
                if v == self.coords[m]:
                    mine = g
                    self.layer_ranks[m] = ranks
            self.layer_groups[m] = mine

    def repl(self, m: int) -> int:
        """Replication factor of factor-m chunks = layer size."""
        return _world() // self.grid[m]

    def localize(self, t: SpTensor) -> SpTensor:
        """Select the nonzeros of this rank's grid box and shift every mode
        to chunk-local coordinates (reference mpi_io.c:756-844)."""
        mask = torch.ones(t.nnz, dtype=torch.bool)
        for m in range(t.nmodes):
            mask &= (t.inds[m] >= self.chunk0[m]) & \
                    (t.inds[m] < self.chunk0[m] + self.chunkn[m])
        inds = t.inds[:, mask].clone()
        for m in range(t.nmodes):
            inds[m] -= self.chunk0[m]
        return SpTensor(inds, t.vals[mask].clone(), list(self.chunkn))


@dataclass
class FineDecomp(GridDecomp):
    """grid == [1]*nm but world > 1: factors fully replicated, partial
    MTTKRP rows reduced across the world. repl() reflects that."""
    nnz_part: Optional[torch.Tensor] = None

    def repl(self, m: int) -> int:
        return _world()

    def localize(self, t: SpTensor) -> SpTensor:
        world = _world()
        if self.nnz_part is not None:
            mask = self.nnz_part == self.rank
        else:
            lo, n = chunk_range(t.nnz, world, self.rank)
            mask = torch.zeros(t.nnz, dtype=torch.bool)
            mask[lo: lo + n] = True
        return SpTensor(t.inds[:, mask].clone(), t.vals[mask].clone(),
                        list(t.dims))


def _ar(t: torch.Tensor, group=None, op=None) -> None:
    if _world() > 1 and group != "solo":
        dist.all_reduce(t, op=op or dist.ReduceOp.SUM, group=group)


def _backend_is_nccl() -> bool:
    """True -> use the real reduce_scatter_tensor / all_gather_into_tensor
    primitives (the RCCL path). SPLATT_FORCE_RS_PRIMS=1 forces this branch
    under gloo (torch>=2.10 implements both on gloo) so CPU tests cover the
    exact padding/ownership code the 8-GPU RCCL run executes."""
    import os
    if os.environ.get("SPLATT_FORCE_RS_PRIMS") == "1":
        return True
    try:
        return dist.get_backend() == "nccl"
    except Exception:  # noqa: BLE001
        return False


def _reduce_scatter_rows(full: torch.Tensor, lo: int, hi: int,
                         group, gsize: int):
    """Sum `full` across the group and return (owned_rows, work) for this
    rank's block [lo, hi); `work` is an async handle (wait before reading
    owned_rows). RCCL: true ncclReduceScatter over equal padded chunks
    (SURVEY.md §2.4 mapping); gloo (CPU tests): all-reduce + slice, which
    is the same math in a different summation order."""
    n, F = full.shape
    if not _backend_is_nccl():
        work = dist.all_reduce(full, group=group, async_op=True)
        return full[lo:hi], work
    per = (n + gsize - 1) // gsize
    src = full.contiguous()
    if n != per * gsize:
        src = torch.zeros(per * gsize, F, dtype=full.dtype,
                          device=full.device)
        src[:n] = full
    out = torch.empty(per, F, dtype=full.dtype, device=full.device)
    work = dist.reduce_scatter_tensor(out, src, group=group, async_op=True)
    return out[: hi - lo], work


def _ag_rows_start(own: torch.Tensor, n: int, group, gsize: int):
    """Begin an async all-gather of the group's owned row blocks; returns
    a handle for _ag_rows_finish."""
    per = (n + gsize - 1) // gsize
    F = own.shape[1]
    src = own
    if own.shape[0] != per:
        src = torch.zeros(per, F, dtype=own.dtype, device=own.device)
        src[: own.shape[0]] = own
    if _backend_is_nccl():
        buf = torch.empty(per * gsize, F, dtype=own.dtype, device=own.device)
        work = dist.all_gather_into_tensor(buf, src.contiguous(),
                                           group=group, async_op=True)
        return (work, buf, None)
    parts = [torch.empty(per, F, dtype=own.dtype, device=own.device)
             for _ in range(gsize)]
    work = dist.all_gather(parts, src.contiguous(), group=group,
                           async_op=True)
    return (work, None, parts)


def _ag_rows_finish(handle, n: int, out: torch.Tensor) -> None:
    work, buf, parts = handle
    if work is not None:
        work.wait()
    if buf is None:
        buf = torch.cat(parts, 0)
    out.copy_(buf[:n])


def _all_gather_rows(own: torch.Tensor, n: int, group, gsize: int,
                     out: torch.Tensor) -> None:
    """Concatenate the group's owned row blocks back into `out` (n x F)."""
    _ag_rows_finish(_ag_rows_start(own, n, group, gsize), n, out)


def _comm_chunks(default: int = 4) -> int:
    """Pipeline depth for the chunked MTTKRP -> reduce-scatter overlap
    (SPLATT_COMM_CHUNKS; 1 disables)."""
    import os
    try:
        return max(1, int(os.environ.get("SPLATT_COMM_CHUNKS", default)))
    except ValueError:
        return default


@dataclass
class GridCpdState:
    cs: CsfSet
    dec: GridDecomp
    factors: List[torch.Tensor]
    grams: List[torch.Tensor]
    lam: torch.Tensor
    buf: torch.Tensor
    norm_x: float
    # reduced-precision factor-store copies for the MTTKRP gathers
    # (SPLATT_FACTOR_STORE)
    qfactors: Optional[List[torch.Tensor]] = None
    fit: float = 0.0
    old_fit: float = 0.0
    niters: int = 0


def grid_cpd_init(shard_cs: CsfSet, dec: GridDecomp, rank_f: int,
                  opts: CpdOptions) -> GridCpdState:
    nm = len(dec.global_dims)
    dev = shard_cs.csfs[0].device
    dtype = shard_cs.csfs[0].vals.dtype

    factors = [seeded_init(dec.chunkn[m], rank_f, m, opts.seed,
                           row0=dec.chunk0[m], dtype=dtype).to(dev)
               for m in range(nm)]
    grams = []
    for m in range(nm):
        g = gram(factors[m])
        if dec.grid[m] > 1:
            g /= dec.repl(m)
            _ar(g)
        grams.append(g)

    nx = torch.tensor([float(shard_cs.csfs[0].vals.double().square().sum())],
                      dtype=torch.float64, device=dev)
    _ar(nx)
    maxdim = max(dec.chunkn)
    from splatt_amd.mttkrp import factor_store_dtype
    qdt = factor_store_dtype() if (dev.type == "cuda"
                                   and dtype == torch.float64) else None
    return GridCpdState(
        cs=shard_cs, dec=dec, factors=factors, grams=grams,
        lam=torch.ones(rank_f, dtype=dtype, device=dev),
        buf=torch.empty(maxdim, rank_f, dtype=dtype, device=dev),
        qfactors=[f.to(qdt) for f in factors] if qdt else None,
        norm_x=float(nx.item()))


def grid_cpd_step(st: GridCpdState, it: int, overlap: bool = True,
                  timers=None) -> float:
    """One ALS iteration. `overlap`: start the mode-m partial-row
    all-reduce asynchronously and form the Gram/Cholesky inverse (which
    do not depend on it) underneath — the comm/compute overlap the
    reference lacks (SURVEY.md §2.4 notes comm is fully synchronous
    there). `timers`: an optional TimerRegistry (sync_device=True for
    honest per-phase times) mirroring the reference's TIMER_MTTKRP /
    TIMER_MPI_* sections."""
    from contextlib import nullcontext
    tm = timers.time if timers is not None else (lambda *_: nullcontext())
    dec = st.dec
    nm = len(dec.global_dims)
    dev = st.buf.device
    dtype = st.buf.dtype
    F = st.factors[0].shape[1]

    import os as _os
    use_rsag = _os.environ.get("SPLATT_NO_RSAG") != "1"
    for m in range(nm):
        mb = st.buf[: dec.chunkn[m]]
        group = dec.layer_groups.get(m)
        distributed = _world() > 1 and group != "solo" and dec.repl(m) > 1
        nrows = dec.chunkn[m]
        # chunked pipeline depth: C partial MTTKRPs by output-row range,
        # each chunk's collective issued as soon as its rows are done and
        # overlapped with the next chunk's MTTKRP (the side-stream overlap
        # the reference lacks — its loop is fully synchronous,
        # mpi/mpi_cpd.c:704-748). RCCL runs the collective on its own HIP
        # stream ordered after the producing kernel.
        C = 1
        if distributed and overlap:
            C = _comm_chunks()
            # pipelining only pays when the exchange is heavy: below
            # SPLATT_COMM_CHUNK_MIN_MB (default 8) a single collective
            # beats C x the collective latency
            import os as _os2
            min_mb = float(_os2.environ.get("SPLATT_COMM_CHUNK_MIN_MB",
                                            "8"))
            payload_mb = nrows * F * st.buf.element_size() / 2**20
            if C > 1 and (nrows < C * dec.repl(m)
                          or payload_mb < min_mb
                          or not mttkrp_rows_ok(st.cs, m, F)):
                C = 1
        bounds = [i * nrows // C for i in range(C + 1)]
        if distributed and use_rsag:
            # the SURVEY §2.4 mapping: reduce-scatter partial rows to
            # contiguous owners (per chunk), solve/normalize/gram ONLY
            # owned rows, all-gather the updated blocks (replaces the
            # reference's alltoallv pair). Gram Hadamard + SPD inverse
            # also run UNDER the async collectives.
            gsize = dec.repl(m)
            my = dec.layer_ranks[m].index(dist.get_rank())
            owns, rs_works = [], []
            for i in range(C):
                clo, chi = bounds[i], bounds[i + 1]
                with tm("MTTKRP"):
                    mttkrp(st.cs, st.qfactors or st.factors, m, out=mb,
                           rows=(clo, chi) if C > 1 else None)
                ni = chi - clo
                per = (ni + gsize - 1) // gsize
                olo = min(my * per, ni)
                ohi = min(olo + per, ni)
                with tm("COMM-RS"):
                    own_i, w = _reduce_scatter_rows(mb[clo:chi], olo, ohi,
                                                    group, gsize)
                owns.append(own_i)
                rs_works.append(w)
            with tm("SOLVE"):
                # fused Hadamard across the other modes (2 launches
                # instead of nm-1 in the eager world>1 tail)
                G = torch.stack([st.grams[o] for o in range(nm)
                                 if o != m]).prod(dim=0)
                Ginv = spd_inverse(G)
            # per-chunk: as each reduce-scatter lands, solve its rows and
            # START its all-gather immediately with UNSCALED rows — the
            # column scale (lambda) is identical on every rank, so
            # dividing after the gather is algebraically the same update;
            # this hides the all-gathers under the remaining solves and
            # the whole lambda/gram section
            solved, ag = [], []
            for i in range(C):
                with tm("COMM-WAIT"):
                    if rs_works[i] is not None:
                        rs_works[i].wait()
                with tm("SOLVE"):
                    Ai = solve_rows(owns[i], Ginv)
                solved.append(Ai)
                with tm("COMM-AG"):
                    ag.append(_ag_rows_start(Ai, bounds[i + 1] - bounds[i],
                                             group, gsize))
            own_mb = owns[0] if C == 1 else torch.cat(owns, 0)
            A_own = solved[0] if C == 1 else torch.cat(solved, 0)
            # lambda over GLOBAL rows (owned rows are globally unique)
            if it == 0:
                s = A_own.square().sum(dim=0)
                _ar(s)
                lam = s.sqrt()
            else:
                lam = A_own.abs().amax(dim=0) if A_own.numel() else                     torch.zeros(F, dtype=dtype, device=dev)
                _ar(lam, op=dist.ReduceOp.MAX)
                lam = lam.clamp_(min=1.0)
            lam = torch.where(lam == 0, torch.ones_like(lam), lam)
            A_own = A_own / lam
            g = gram(A_own) if A_own.numel() else                 torch.zeros(F, F, dtype=dtype, device=dev)
            _ar(g)
            if m == nm - 1:
                # fit inner from OWNED rows (each global row owned exactly
                # once across the job): partial now, summed at fit time
                part = (own_mb.double() * A_own.double()).sum(dim=0)                     if A_own.numel() else torch.zeros(F, dtype=torch.float64,
                                                      device=dev)
                st._rs_inner = part  # type: ignore[attr-defined]
            A = torch.empty(nrows, F, dtype=dtype, device=dev)
            with tm("COMM-AG"):
                for i in range(C):
                    _ag_rows_finish(ag[i], bounds[i + 1] - bounds[i],
                                    A[bounds[i]: bounds[i + 1]])
            A /= lam
            st.lam = lam
            st.factors[m] = A
            if st.qfactors is not None:
                st.qfactors[m] = A.to(st.qfactors[m].dtype)
            st.grams[m] = g
            continue
        works = []
        if distributed:
            for i in range(C):
                clo, chi = bounds[i], bounds[i + 1]
                with tm("MTTKRP"):
                    mttkrp(st.cs, st.qfactors or st.factors, m, out=mb,
                           rows=(clo, chi) if C > 1 else None)
                with tm("COMM-POST"):
                    works.append(dist.all_reduce(mb[clo:chi], group=group,
                                                 async_op=True))
        else:
            with tm("MTTKRP"):
                mttkrp(st.cs, st.qfactors or st.factors, m, out=mb)
        with tm("SOLVE"):
            G = torch.stack([st.grams[o] for o in range(nm)
                             if o != m]).prod(dim=0)
            Ginv = spd_inverse(G)
        with tm("COMM-WAIT"):
            for w in works:
                if w is not None:
                    w.wait()
        with tm("SOLVE"):
            A = solve_rows(mb, Ginv)
        # lambda over GLOBAL rows of mode m
        if it == 0:
            s = A.square().sum(dim=0)
            if dec.grid[m] > 1:
                s /= dec.repl(m)
                _ar(s)
            lam = s.sqrt()
        else:
            lam = A.abs().amax(dim=0)
            if dec.grid[m] > 1:
                _ar(lam, op=dist.ReduceOp.MAX)
            lam = lam.clamp_(min=1.0)
        lam = torch.where(lam == 0, torch.ones_like(lam), lam)
        A /= lam
        st.lam = lam
        st.factors[m] = A
        if st.qfactors is not None:
            st.qfactors[m] = A.to(st.qfactors[m].dtype)
        g = gram(A)
        if dec.grid[m] > 1:
            g /= dec.repl(m)
            _ar(g)
        st.grams[m] = g

    mlast = nm - 1
    rs_part = getattr(st, "_rs_inner", None)
    if rs_part is not None:
        _ar(rs_part)
        inner_t = (rs_part * st.lam.double()).sum()
        st._rs_inner = None  # type: ignore[attr-defined]
    else:
        inner_t = ((st.buf[: dec.chunkn[mlast]].double()
                    * st.factors[mlast].double()).sum(dim=0)
                   * st.lam.double()).sum()
        if dec.grid[mlast] > 1:
            inner_t /= dec.repl(mlast)
            _ar(inner_t)
    inner = float(inner_t)
    Gall = torch.ones(F, F, dtype=dtype, device=dev)
    for o in range(nm):
        Gall *= st.grams[o]
    knorm = float((Gall.double()
                   * torch.outer(st.lam.double(), st.lam.double())).sum())
    residual = math.sqrt(max(0.0, st.norm_x + knorm - 2 * inner))
    st.old_fit = st.fit
    st.fit = 1.0 - residual / math.sqrt(st.norm_x)
    st.niters = it + 1
    return st.fit


def comm_stats(dec: GridDecomp, shard_nnz: int, rank_f: int,
               val_bytes: int = 8, cs: Optional[CsfSet] = None) -> dict:
    """Per-rank communication volume per ALS iteration + nnz balance
    (reference mpi_rank_stats / mpi_cpd_stats, stats.c:298-465).
    COLLECTIVE: every rank must call this (it all-reduces nnz counts).
    With `cs`, also reports rows actually TOUCHED by local nonzeros vs
    rows EXCHANGED (the whole contiguous chunk): the contiguous-chunk
    RS/AG trades need-based lists (reference ineed, mpi_setup.c:13-155)
    for true reduce-scatter — this keeps that trade measured."""
    world = _world()
    per_mode = []
    total = 0
    touched = []
    for m in range(len(dec.global_dims)):
        if world > 1 and dec.repl(m) > 1:
            # ring all-reduce over the layer: ~2x payload per member
            payload = dec.chunkn[m] * rank_f * val_bytes
            vol = 2 * payload * (dec.repl(m) - 1) // max(dec.repl(m), 1)
        else:
            vol = 0
        per_mode.append(vol)
        total += vol
        if cs is not None:
            c = cs.csfs[cs.mode_csf[m]]
            lab = c.ancestor_expand(cs.mode_depth[m])
            tr = torch.tensor([float(torch.unique(lab).numel())],
                              dtype=torch.float64, device=lab.device)
            if world > 1:
                dist.all_reduce(tr, op=dist.ReduceOp.MAX)
            touched.append(int(tr.item()))
    nnz_t = torch.tensor([float(shard_nnz)], dtype=torch.float64)
    if world > 1:
        mx = nnz_t.clone()
        dist.all_reduce(mx, op=dist.ReduceOp.MAX)
        sm = nnz_t.clone()
        dist.all_reduce(sm)
        imbalance = float(mx) / (float(sm) / world) - 1.0
    else:
        imbalance = 0.0
    out = {"grid": dec.grid, "comm_bytes_per_iter": total,
           "comm_bytes_per_mode": per_mode,
           "nnz_imbalance": round(imbalance, 4)}
    if cs is not None:
        out["touched_rows_per_mode"] = touched
        out["exchanged_rows_per_mode"] = [
            dec.chunkn[m] if per_mode[m] else 0
            for m in range(len(dec.global_dims))]
    return out


def grid_cpd_als(shard_cs: CsfSet, dec: GridDecomp, rank_f: int,
                 opts: Optional[CpdOptions] = None) -> Kruskal:
    opts = opts or CpdOptions()
    st = grid_cpd_init(shard_cs, dec, rank_f, opts)
    trace = []
    import time as _time
    for it in range(opts.max_iters):
        _t0 = _time.perf_counter()
        fit = grid_cpd_step(st, it)
        trace.append(fit)
        if opts.verbose and (_world() == 1 or dist.get_rank() == 0):
            print(f"  its = {it + 1} ({_time.perf_counter() - _t0:.3f}s) "
                  f"fit = {fit:.5f}", flush=True)
        if it > 0 and abs(fit - st.old_fit) < opts.tolerance:
            break
    # post-process with GLOBAL column norms (chunked factors)
    for m, A in enumerate(st.factors):
        s2 = A.square().sum(dim=0)
        if dec.grid[m] > 1:
            s2 /= dec.repl(m)
            _ar(s2)
        norms = s2.sqrt()
        norms = torch.where(norms == 0, torch.ones_like(norms), norms)
        A /= norms
        st.lam *= norms
    return Kruskal(factors=st.factors, lam=st.lam, fit=st.fit,
                   niters=st.niters, fit_trace=trace)


def load_shard(path: str, dec: GridDecomp, dtype=None):
    """Distributed tensor load: every rank reads the (shared-filesystem)
    file and keeps its box (single-node analog of mpi_tt_read,
    mpi/mpi_io.c:756; chunked root->rank streaming is unnecessary when all
    ranks share one node's filesystem)."""
    import torch as _torch
    t = SpTensor.load(path, dtype or _torch.float64)
    return dec.localize(t)


def gather_factors(k: Kruskal, dec: GridDecomp):
    """Assemble full (global-row) factor matrices on rank 0 from chunked
    factors (reference mpi_write_mats root-gather, mpi/mpi_io.c:927)."""
    world = _world()
    if world == 1:
        return list(k.factors)
    rank = dist.get_rank()
    full = []
    for m, A in enumerate(k.factors):
        if dec.grid[m] == 1:
            full.append(A if rank == 0 else None)
            continue
        pieces = [None] * world
        dist.all_gather_object(pieces, (dec.coords[m], dec.chunk0[m],
                                        A.cpu()))
        if rank == 0:
            seen = {}
            for coord, c0, chunk in pieces:
                seen[c0] = chunk   # layer replicas collapse
            rows = [seen[c0] for c0 in sorted(seen)]
            full.append(__import__("torch").cat(rows, dim=0))
        else:
            full.append(None)
    return full


def write_factors(k: Kruskal, dec: GridDecomp, prefix: str = "") -> None:
    """Rank 0 writes modeN.mat + lambda.mat with GLOBAL rows."""
    full = gather_factors(k, dec)
    if _world() > 1 and dist.get_rank() != 0:
        return
    for m, A in enumerate(full):
        with open(f"{prefix}mode{m + 1}.mat", "w") as fh:
            for row in A.cpu().tolist():
                fh.write(" ".join(f"{x:.17g}" for x in row) + "\n")
    with open(f"{prefix}lambda.mat", "w") as fh:
        for x in k.lam.cpu().tolist():
            fh.write(f"{x:.17g}\n")
