"""CPD-ALS drivers.

Capability parity: reference src/cpd.c (cpd_als_iterate:271-387, fit
math:116-265, normalize schedule cpd.c:343-347, post-process:391-411).

Two drivers:
  * `cpd_als` — the device-resident driver: HIP MTTKRP + rocBLAS (via torch)
    for Gram/solve/normalize. Factors stay in HBM for the whole run; only
    lambda and the fit scalar cross to host. Also runs on CPU tensors (torch
    CPU + C++ CSF MTTKRP), which is what the gloo multi-process tests use.
  * `cpd_als_cpu_native` — the pure C++ reference path (`splatt cpd`
    1-thread CPU config of BASELINE.json).
"""
from __future__ import annotations

import math
from dataclasses import dataclass, field
from typing import List, Optional

import torch

from splatt_amd._ext import native
from splatt_amd.csf import CsfSet, csf_alloc
from splatt_amd.mttkrp import mttkrp
from splatt_amd.ops.dense import gram, solve_rows, spd_inverse
from splatt_amd.sptensor import SpTensor


@dataclass
class CpdOptions:
    tolerance: float = 1e-5
    max_iters: int = 50
    seed: int = 0x5EED5EED
    csf_alloc: str = "two"
    nthreads: int = 0
    verbose: bool = False
    regularize: float = 0.0   # ridge term on the Gram diagonal
                              # (reference p_form_gram + reg*I)
    # per-iteration factor checkpointing (the reference has none —
    # SURVEY.md §5 flags this as a cheap rebuild improvement)
    checkpoint_path: str = ""
    checkpoint_every: int = 1
    resume: bool = False


@dataclass
class Kruskal:
    factors: List[torch.Tensor]
    lam: torch.Tensor
    fit: float
    niters: int
    fit_trace: List[float] = field(default_factory=list)


def seeded_init(nrows: int, rank: int, mode: int, seed: int,
                row0: int = 0, dtype: torch.dtype = torch.float64) -> torch.Tensor:
    """Partition-invariant factor init: element (i,f) depends only on
    (seed, mode, global row i, f) — the property the reference gets from
    root-generated mpi_mat_rand (mpi/mpi_io.c:1097-1176)."""
    tag = "f32" if dtype == torch.float32 else "f64"
    return native().seeded_init(nrows, rank, row0, seed, mode, tag)


def _normalize(A: torch.Tensor, it: int) -> torch.Tensor:
    if it == 0:
        lam = A.square().sum(dim=0).sqrt()
    else:
        lam = A.abs().amax(dim=0).clamp_(min=1.0)
    lam = torch.where(lam == 0, torch.ones_like(lam), lam)
    A /= lam
    return lam


def cpd_als(src: CsfSet | SpTensor, rank: int,
            opts: Optional[CpdOptions] = None) -> Kruskal:
    opts = opts or CpdOptions()
    cs = src if isinstance(src, CsfSet) else csf_alloc(src, opts.csf_alloc)
    nm = cs.nmodes
    dims = cs.dims
    dev = cs.csfs[0].device
    dtype = cs.csfs[0].vals.dtype

    factors = [seeded_init(dims[m], rank, m, opts.seed, dtype=dtype).to(dev)
               for m in range(nm)]
    grams = [gram(f) for f in factors]
    from splatt_amd.mttkrp import factor_store_dtype
    qdt = factor_store_dtype() if (dev.type == "cuda"
                                   and dtype == torch.float64) else None
    qfactors = [f.to(qdt) for f in factors] if qdt else None
    norm_x = float(cs.csfs[0].vals.double().square().sum())
    lam = torch.ones(rank, dtype=dtype, device=dev)
    buf = torch.empty(max(dims), rank, dtype=dtype, device=dev)

    fit = old_fit = 0.0
    trace: List[float] = []
    niters = 0
    ones = torch.ones(rank, rank, dtype=dtype, device=dev)

    it0 = 0
    if opts.resume and opts.checkpoint_path:
        import os
        if os.path.exists(opts.checkpoint_path):
            ck = torch.load(opts.checkpoint_path, map_location=dev,
                            weights_only=True)
            meta = ck.get("meta")
            want = {"dims": list(dims), "rank": rank, "dtype": str(dtype),
                    "seed": opts.seed}
            if meta is not None and meta != want:
                bad = {k: (meta.get(k), want[k]) for k in want
                       if meta.get(k) != want[k]}
                raise ValueError(
                    f"checkpoint {opts.checkpoint_path} does not match this "
                    f"run (checkpoint vs current): {bad}")
            factors = [f.to(dev) for f in ck["factors"]]
            grams = [gram(f) for f in factors]
            if qfactors is not None:
                qfactors = [f.to(qdt) for f in factors]
            lam = ck["lambda"].to(dev)
            it0 = int(ck["iteration"]) + 1
            fit = old_fit = float(ck["fit"])
            trace = list(ck.get("fit_trace", []))

    import time as _time
    for it in range(it0, opts.max_iters):
        _t0 = _time.perf_counter()
        for m in range(nm):
            mb = buf[: dims[m]]
            mttkrp(cs, qfactors or factors, m, out=mb,
                   nthreads=opts.nthreads)
            G = ones.clone()
            for o in range(nm):
                if o != m:
                    G *= grams[o]
            if opts.regularize:
                G += opts.regularize * torch.eye(rank, dtype=dtype,
                                                 device=dev)
            # solve A * G = mttkrp  =>  G^T A^T = mttkrp^T (G symmetric)
            # F x F inverse once, then one well-shaped (n x F)(F x F) GEMM —
            # beats a trsm against an n-row RHS at these tiny F
            A = solve_rows(mb, spd_inverse(G))
            lam = _normalize(A, it)
            factors[m] = A
            if qfactors is not None:
                qfactors[m] = A.to(qdt)
            grams[m] = gram(A)

        # fit from last mode's pre-solve MTTKRP output (reference trick):
        # <X,K> = sum_f lam_f * sum_i buf[i,f] * A_last[i,f]
        mlast = nm - 1
        inner = float(((buf[: dims[mlast]].double()
                        * factors[mlast].double()).sum(dim=0)
                       * lam.double()).sum())
        Gall = ones.clone()
        for o in range(nm):
            Gall *= grams[o]
        knorm = float((Gall.double()
                       * torch.outer(lam.double(), lam.double())).sum())
        residual = math.sqrt(max(0.0, norm_x + knorm - 2 * inner))
        fit = 1.0 - residual / math.sqrt(norm_x)
        trace.append(fit)
        niters = it + 1
        if opts.verbose:
            print(f"  its = {it + 1} ({_time.perf_counter() - _t0:.3f}s) "
                  f"fit = {fit:.5f} delta = {fit - old_fit:+.4e}")
        if opts.checkpoint_path and (it + 1) % opts.checkpoint_every == 0:
            import os
            tmp = opts.checkpoint_path + ".tmp"
            torch.save({"factors": [f.cpu() for f in factors],
                        "lambda": lam.cpu(), "iteration": it, "fit": fit,
                        "fit_trace": trace,
                        "meta": {"dims": list(dims), "rank": rank,
                                 "dtype": str(dtype), "seed": opts.seed}},
                       tmp)
            os.replace(tmp, opts.checkpoint_path)
        if it > 0 and abs(fit - old_fit) < opts.tolerance:
            break
        old_fit = fit

    _post_process(factors, lam)
    return Kruskal(factors=factors, lam=lam, fit=fit, niters=niters,
                   fit_trace=trace)


def _post_process(factors, lam) -> None:
    """Final renormalization: unit 2-norm columns, scales folded into
    lambda (reference cpd_post_process, cpd.c:391-411)."""
    for A in factors:
        norms = A.square().sum(dim=0).sqrt()
        norms = torch.where(norms == 0, torch.ones_like(norms), norms)
        A /= norms
        lam *= norms


def cpd_als_cpu_native(t: SpTensor, rank: int,
                       opts: Optional[CpdOptions] = None) -> Kruskal:
    """The C++ host reference path (BASELINE config 1: 1-thread CPU cpd)."""
    opts = opts or CpdOptions()
    r = native().cpd_als_cpu(
        t.inds.cpu(), t.vals.cpu(), list(t.dims), rank,
        {
            "tolerance": opts.tolerance,
            "max_iters": opts.max_iters,
            "seed": opts.seed,
            "csf_alloc": opts.csf_alloc,
            "nthreads": opts.nthreads,
            "regularize": opts.regularize,
        })
    return Kruskal(factors=list(r["factors"]), lam=r["lambda"],
                   fit=float(r["fit"]), niters=int(r["niters"]))
