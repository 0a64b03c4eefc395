"""MTTKRP algorithm benchmark harness.

Capability parity: reference src/bench.c + cmds/cmd_bench.c — `splatt
bench -a {splatt,csf,giga,ttbox,coord}` times each algorithm per mode per
iteration with optional cross-validation against the gold result. Our
algorithm set: 'stream' (COO oracle), 'csf' (hierarchical fiber-walk HIP
kernels / C++ CPU walker), 'flat' (staged expanded-CSF HIP kernel — the
production path).
"""
from __future__ import annotations

import time
from typing import Dict, List, Optional

import torch

from splatt_amd.cpd import seeded_init
from splatt_amd.csf import csf_alloc
from splatt_amd.mttkrp import mttkrp, mttkrp_stream
from splatt_amd.sptensor import SpTensor

ALGS = ("flat", "csf", "stream")
LEGACY_ALGS = ("giga", "ttbox")  # reference benchmark baselines


def _mttkrp_giga(t, mats, mode):
    """GigaTensor-style: sparse unfolding X_(m) times the materialized
    Khatri-Rao product (reference mttkrp_giga, mttkrp.c:1604-1649)."""
    others = [m for m in range(t.nmodes) if m != mode]
    X = t.unfold(mode)
    K = mats[others[0]]
    for m in others[1:]:
        K = (K.unsqueeze(1) * mats[m].unsqueeze(0)).reshape(-1, K.shape[1])
    return torch.sparse.mm(X, K)


def _mttkrp_ttbox(t, mats, mode):
    """Tensor-Toolbox style: one output column at a time via elementwise
    products + index_add (reference mttkrp_ttbox, mttkrp.c:1655-1695)."""
    rank = mats[0].shape[1]
    out = torch.zeros(t.dims[mode], rank, dtype=t.vals.dtype, device=t.device)
    for f in range(rank):
        w = t.vals.clone()
        for m in range(t.nmodes):
            if m != mode:
                w *= mats[m][t.inds[m], f]
        out[:, f].index_add_(0, t.inds[mode], w)
    return out


def bench_mttkrp(t: SpTensor, rank: int, algs: List[str] = None,
                 niters: int = 3, device: str = "cpu",
                 policy: str = "all", validate: bool = False,
                 seed: int = 123,
                 threads: List[int] = None) -> Dict[str, dict]:
    """Time every algorithm for every mode; returns per-alg results with
    seconds per mode and effective GFLOP/s (3*nnz*rank flops/MTTKRP).
    `threads`: CPU thread counts to sweep for the CSF algorithms — the
    reference bench's thread-scaling mode (bench.c --threads)."""
    algs = list(algs or ALGS)
    dev = torch.device(device)
    mats = [seeded_init(d, rank, m, seed).to(dev)
            for m, d in enumerate(t.dims)]
    td = t.to(dev)
    cs = csf_alloc(td, policy)
    # "lds": the production staged/packed device build (what bench.py
    # and the CPD drivers run) as a benchmarkable algorithm
    cs_lds = None
    if dev.type == "cuda" and rank in (4, 8, 16, 32, 64):
        from splatt_amd.parallel.dist_cpd import build_shard_csf
        cs_lds = build_shard_csf(td, list(t.dims), policy,
                                 flat_only=True, stage_rank=rank)
    elif "lds" in algs:
        algs.remove("lds")
    gold = None
    if validate:
        gold = [mttkrp_stream(t, [m.cpu() for m in mats], mode)
                for mode in range(t.nmodes)]

    def sync():
        if dev.type == "cuda":
            torch.cuda.synchronize()

    results: Dict[str, dict] = {}
    flops = 3.0 * t.nnz * rank
    sweep = [(alg, nt) for alg in algs
             for nt in ((threads or [0]) if (threads and dev.type == "cpu"
                                             and alg in ("flat", "csf"))
                        else [0])]
    for alg, nt in sweep:
        per_mode = []
        ok = True
        for mode in range(t.nmodes):
            def run():
                if alg == "stream":
                    return mttkrp_stream(t, [m.cpu() for m in mats], mode)
                if alg == "giga":
                    return _mttkrp_giga(td, mats, mode)
                if alg == "ttbox":
                    return _mttkrp_ttbox(td, mats, mode)
                if alg == "lds":
                    return mttkrp(cs_lds, mats, mode)
                return mttkrp(cs, mats, mode, alg=alg, nthreads=nt)
            out = run()  # warmup + result for validation
            if gold is not None:
                err = (out.cpu().double() - gold[mode].double()).abs().max()
                ok &= bool(err < 1e-6)
            sync()
            tic = time.perf_counter()
            for _ in range(niters):
                out = run()
            sync()
            per_mode.append((time.perf_counter() - tic) / niters)
        results[alg if nt == 0 else f"{alg}@t{nt}"] = {
            "seconds_per_mode": per_mode,
            "gflops_per_mode": [flops / s / 1e9 for s in per_mode],
            "validated": ok if gold is not None else None,
        }
    return results


def format_bench(results: Dict[str, dict]) -> str:
    lines = []
    for alg, r in results.items():
        secs = " ".join(f"{s * 1e3:9.3f}ms" for s in r["seconds_per_mode"])
        gfs = " ".join(f"{g:8.1f}" for g in r["gflops_per_mode"])
        v = "" if r["validated"] is None else \
            ("  [validated]" if r["validated"] else "  [MISMATCH]")
        lines.append(f"  {alg:8s} per-mode: {secs}   GFLOP/s: {gfs}{v}")
    return "\n".join(lines)
