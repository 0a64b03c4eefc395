"""Command-line interface: `python -m splatt_amd <cmd>` (the native
`bin/splatt` binary covers the host-library surface).

Capability parity: the reference `splatt` binary and its sub-commands
(cmds/splatt_cmds.h:18-27): cpd, bench, check, convert, reorder, stats.
"""
from __future__ import annotations

import argparse
import sys

import torch

import splatt_amd as sp
from splatt_amd.stats import cpd_stats, stats_csf, stats_tt
from splatt_amd.utils.timers import TIMERS


def _add_common(p):
    p.add_argument("tensor", help="input tensor (.tns/.coo text or .bin)")
    p.add_argument("--dtype", default="f64", choices=["f64", "f32"])


def _load(args) -> "sp.SpTensor":
    dtype = torch.float64 if args.dtype == "f64" else torch.float32
    with TIMERS.time("IO"):
        t = sp.SpTensor.load(args.tensor, dtype)
    return t


def cmd_cpd(args) -> int:
    import os
    if getattr(args, "csf", None) is None:
        use_cuda = (args.device in ("auto", "cuda")
                    and torch.cuda.is_available())
        args.csf = "all" if use_cuda else "two"
    if getattr(args, "factor_store", "f64") != "f64":
        os.environ["SPLATT_FACTOR_STORE"] = args.factor_store
    if getattr(args, "deterministic", False):
        # bitwise-reproducible device kernels (docs/KERNELS.md); forces
        # ALLMODE so every mode has a root-sorted stream
        os.environ["SPLATT_DETERMINISTIC"] = "1"
        args.csf = "all"
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world > 1:
        return _cmd_cpd_dist(args, world)
    t = _load(args)
    print(stats_tt(t, args.tensor))
    opts = sp.CpdOptions(tolerance=args.tol, max_iters=args.its,
                         seed=args.seed, csf_alloc=args.csf,
                         nthreads=args.nthreads, verbose=args.verbose > 0,
                         regularize=args.reg)
    dev = args.device
    if dev == "auto":
        dev = "cuda" if torch.cuda.is_available() else "cpu"
    with TIMERS.time("CPD"):
        if dev == "cpu" and args.native:
            k = sp.cpd_als_cpu_native(t, args.rank, opts)
        elif dev == "cuda":
            # production device build: flat streams + LDS-staged buckets
            # (the same path bench.py and the distributed driver use)
            from splatt_amd.parallel.dist_cpd import build_shard_csf
            stage = args.rank if args.rank in (4, 8, 16, 32, 64) else 0
            cs = build_shard_csf(t.to(dev), list(t.dims), args.csf,
                                 flat_only=True, stage_rank=stage)
            print(stats_csf(cs))
            print(cpd_stats(cs, args.rank, opts))
            k = sp.cpd_als(cs, args.rank, opts)
        else:
            cs = sp.csf_alloc(t.to(dev), args.csf)
            print(stats_csf(cs))
            print(cpd_stats(cs, args.rank, opts))
            k = sp.cpd_als(cs, args.rank, opts)
    print(f"Final fit: {k.fit:.5f}  (iterations: {k.niters})")
    if not args.nowrite:
        for m, f in enumerate(k.factors):
            with open(f"mode{m + 1}.mat", "w") as fh:
                for row in f.cpu().tolist():
                    fh.write(" ".join(f"{x:.17g}" for x in row) + "\n")
        with open("lambda.mat", "w") as fh:
            for x in k.lam.cpu().tolist():
                fh.write(f"{x:.17g}\n")
    print(TIMERS.report())
    return 0


def _cmd_cpd_dist(args, world: int) -> int:
    """`torchrun --nproc-per-node N -m splatt_amd cpd ...` — the analog of
    `mpirun splatt cpd` (reference cmds/mpi_cmd_cpd.c:175): grid
    decomposition over RCCL (GPU) or gloo (CPU), rank-0 output."""
    import os
    import torch.distributed as dist
    from splatt_amd.parallel.dist_cpd import build_shard_csf
    from splatt_amd.parallel.grid import (GridDecomp, grid_cpd_als,
                                          load_shard, write_factors)
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    use_cuda = args.device in ("auto", "cuda") and torch.cuda.is_available()
    if use_cuda:
        torch.cuda.set_device(local_rank % torch.cuda.device_count())
    dist.init_process_group("nccl" if use_cuda else "gloo")
    dtype = torch.float64 if args.dtype == "f64" else torch.float32
    t = sp.SpTensor.load(args.tensor, dtype)
    dec = GridDecomp.create(list(t.dims))
    shard = dec.localize(t)
    if use_cuda:
        shard = shard.to(f"cuda:{local_rank % torch.cuda.device_count()}")
    stage = args.rank if (use_cuda
                          and args.rank in (4, 8, 16, 32, 64)) else 0
    cs = build_shard_csf(shard, list(t.dims), args.csf,
                         flat_only=use_cuda, stage_rank=stage)
    opts = sp.CpdOptions(tolerance=args.tol, max_iters=args.its,
                         seed=args.seed, csf_alloc=args.csf)
    k = grid_cpd_als(cs, dec, args.rank, opts)
    if rank == 0:
        print(f"Final fit: {k.fit:.5f}  (iterations: {k.niters}; "
              f"grid {dec.grid})")
    if not args.nowrite:
        write_factors(k, dec)
    dist.destroy_process_group()
    return 0


def cmd_check(args) -> int:
    t = _load(args)
    fixed = t.fixed(dedup=True, compress=args.compress)
    ndups = getattr(fixed, "_ndups", 0)
    nempty = getattr(fixed, "_nempty", 0)
    print(f"duplicates merged: {ndups}; empty slices removed: {nempty}")
    if args.fix:
        fixed.save(args.fix)
        print(f"wrote {args.fix}")
    return 0


def cmd_convert(args) -> int:
    t = _load(args)
    kind = args.type
    if kind == "auto":
        kind = "bin" if args.output.endswith(".bin") else "tns"
    if kind in ("bin", "tns"):
        if kind == "bin" and not args.output.endswith(".bin"):
            args.output += ".bin"
        t.save(args.output)
    elif kind == "graph":
        from splatt_amd.graph import graph_mpartite, graph_write
        graph_write(graph_mpartite(t), args.output)
    elif kind == "hgraph":
        from splatt_amd.graph import hgraph_nnz, hgraph_write
        hgraph_write(hgraph_nnz(t), args.output)
    elif kind == "fib_hgraph":
        from splatt_amd.graph import hgraph_fib, hgraph_write
        hgraph_write(hgraph_fib(t, args.mode), args.output)
    elif kind == "csr":
        # mode-`--mode` CSR unfolding (reference CNV_FIB_SPMAT,
        # convert.c:134): "nrows ncols nnz" header then one
        # "col val" pair list per row
        X = t.unfold(args.mode)
        cp = X.crow_indices().tolist()
        ci = X.col_indices().tolist()
        v = X.values().tolist()
        with open(args.output, "w") as f:
            f.write(f"{X.shape[0]} {X.shape[1]} {len(v)}\n")
            for r in range(X.shape[0]):
                f.write(" ".join(f"{ci[i] + 1} {v[i]:g}"
                                 for i in range(cp[r], cp[r + 1])) + "\n")
    print(f"wrote {args.output} ({kind})")
    return 0


def cmd_stats(args) -> int:
    t = _load(args)
    print(stats_tt(t, args.tensor))
    if getattr(args, "part", None):
        from splatt_amd.graph import part_read
        from splatt_amd.stats import stats_hparts
        print(stats_hparts(t, part_read(args.part), args.part))
        return 0
    cs = sp.csf_alloc(t, args.csf)
    print(stats_csf(cs))
    return 0


def cmd_bench(args) -> int:
    from splatt_amd.benchmarks import bench_mttkrp, format_bench
    t = _load(args)
    print(stats_tt(t, args.tensor))
    dev = args.device
    if dev == "auto":
        dev = "cuda" if torch.cuda.is_available() else "cpu"
    threads = [int(x) for x in args.threads.split(",")] if args.threads \
        else None
    res = bench_mttkrp(t, args.rank, args.algs.split(","), args.iters,
                       device=dev, validate=args.validate, threads=threads)
    print(format_bench(res))
    return 0


def cmd_reorder(args) -> int:
    from splatt_amd import reorder as ro
    from splatt_amd.graph import part_read
    t = _load(args)
    if args.type == "rand":
        perm = ro.perm_rand(t.dims, args.seed)
    elif args.type == "bfs":
        perm = ro.perm_bfs(t)
    elif args.type == "graph":
        perm = ro.perm_graph(t, part_read(args.partfile))
    elif args.type == "hgraph":
        perm = ro.perm_hgraph(t, part_read(args.partfile))
    else:
        raise SystemExit(f"unknown reorder type {args.type}")
    out = ro.perm_apply(t, perm)
    out.save(args.output)
    if args.permfile:
        ro.perm_write(perm, args.permfile)
    print(f"wrote {args.output}")
    return 0


def main(argv=None) -> int:
    ap = argparse.ArgumentParser(
        prog="splatt",
        description="MI355X-native sparse tensor factorization "
                    "(CPD-ALS / CSF MTTKRP)")
    ap.add_argument("--version", action="version", version=sp.__version__)
    sub = ap.add_subparsers(dest="cmd", required=True)

    p = sub.add_parser("cpd", help="compute the CPD of a sparse tensor")
    _add_common(p)
    p.add_argument("-r", "--rank", type=int, default=10)
    p.add_argument("-t", "--tol", type=float, default=1e-5)
    p.add_argument("-i", "--its", type=int, default=50)
    p.add_argument("--seed", type=int, default=0x5EED5EED)
    p.add_argument("--reg", type=float, default=0.0,
                   help="ridge regularization on the normal equations")
    p.add_argument("--csf", default=None, choices=["one", "two", "all"],
                   help="CSF allocation policy (default: 'all' on GPU — "
                        "root-output streams for every mode, 3-5x faster "
                        "kernels; 'two' on CPU, the reference default)")
    p.add_argument("--device", default="auto", choices=["auto", "cpu", "cuda"])
    p.add_argument("--nthreads", type=int, default=0)
    p.add_argument("--native", action="store_true",
                   help="use the C++ host driver (CPU reference path)")
    p.add_argument("--nowrite", action="store_true")
    p.add_argument("--factor-store", default="f64",
                   choices=["f64", "f32", "bf16"],
                   help="reduced-precision factor STORAGE for the device "
                        "MTTKRP gathers (accumulation stays f64; "
                        "docs/TUNING.md)")
    p.add_argument("--deterministic", action="store_true",
                   help="bitwise-reproducible device CPD (atomic-free "
                        "kernels, >=80%% throughput; implies --csf all)")
    p.add_argument("-v", "--verbose", action="count", default=0)
    p.set_defaults(fn=cmd_cpd)

    p = sub.add_parser("check", help="find/repair duplicates + empty slices")
    _add_common(p)
    p.add_argument("--fix", help="write the repaired tensor here")
    p.add_argument("--compress", action="store_true",
                   help="also remove empty slices")
    p.set_defaults(fn=cmd_check)

    p = sub.add_parser("convert", help="convert tensor/graph formats")
    _add_common(p)
    p.add_argument("output")
    p.add_argument("-t", "--type", default="auto",
                   choices=["auto", "bin", "tns", "graph", "hgraph",
                            "fib_hgraph", "csr"])
    p.add_argument("-m", "--mode", type=int, default=0,
                   help="mode for fib_hgraph/csr conversions")
    p.set_defaults(fn=cmd_convert)

    p = sub.add_parser("stats", help="print tensor statistics")
    _add_common(p)
    p.add_argument("--part", help="nnz partition file: report partition "
                                  "quality (reference --hparts analysis)")
    p.add_argument("--csf", default="two", choices=["one", "two", "all"])
    p.set_defaults(fn=cmd_stats)

    p = sub.add_parser("bench", help="benchmark MTTKRP algorithms")
    _add_common(p)
    p.add_argument("-r", "--rank", type=int, default=16)
    p.add_argument("-a", "--algs", default="flat,csf,stream",
                   help="comma list from: flat, csf, stream, giga, ttbox, lds (production staged device path)")
    p.add_argument("-N", "--iters", type=int, default=3)
    p.add_argument("--device", default="auto", choices=["auto", "cpu", "cuda"])
    p.add_argument("--validate", action="store_true")
    p.add_argument("--threads", help="comma list of CPU thread counts to "
                                     "sweep (reference bench scaling mode)")
    p.set_defaults(fn=cmd_bench)

    p = sub.add_parser("reorder", help="reorder a tensor")
    _add_common(p)
    p.add_argument("output")
    p.add_argument("--type", default="bfs",
                   choices=["rand", "bfs", "graph", "hgraph"])
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--partfile", help="partition file from an external "
                                      "partitioner (graph/hgraph types)")
    p.add_argument("--permfile", help="prefix to write .modeN.perm files")
    p.set_defaults(fn=cmd_reorder)

    args = ap.parse_args(argv)
    try:
        return args.fn(args)
    except (RuntimeError, OSError, ValueError) as e:
        print(f"splatt: {e}", file=sys.stderr)
        return 1


if __name__ == "__main__":
    sys.exit(main())
