"""Tensor / CSF / CPD statistics reporting.

Capability parity: reference src/stats.c (stats_tt:26-50, stats_csf,
cpd_stats:226-295) and util.c bytes_str humanization.
"""
from __future__ import annotations

from splatt_amd.csf import CsfSet
from splatt_amd.sptensor import SpTensor


def bytes_str(n: float) -> str:
    for unit in ("B", "KB", "MB", "GB", "TB"):
        if n < 1024 or unit == "TB":
            return f"{n:.2f}{unit}"
        n /= 1024.0
    return f"{n:.2f}TB"


def stats_tt(t: SpTensor, name: str = "") -> str:
    s = t.stats()
    dims = "x".join(str(d) for d in s["dims"])
    return (f"Tensor information ---------------------------------\n"
            f"FILE={name or '(in-memory)'}\n"
            f"DIMS={dims} NNZ={s['nnz']} DENSITY={s['density']:.4e}\n"
            f"COORD-STORAGE={bytes_str(s['coo_bytes'])}\n")


def stats_csf(cs: CsfSet) -> str:
    lines = ["CSF information ------------------------------------"]
    for i, c in enumerate(cs.csfs):
        order = "-".join(str(m) for m in c.dim_perm)
        nf = " ".join(str(c.nfibs(l)) for l in range(c.nmodes))
        lines.append(f"CSF-{i}: modes={order} nfibs=[{nf}] "
                     f"storage={bytes_str(c.storage_bytes())}")
    lines.append(f"TOTAL-CSF-STORAGE={bytes_str(cs.storage_bytes())}")
    return "\n".join(lines)


def cpd_stats(cs: CsfSet, rank: int, opts) -> str:
    return (f"Factoring ------------------------------------------\n"
            f"NFACTORS={rank} MAXITS={opts.max_iters} TOL={opts.tolerance:.0e} "
            f"SEED={opts.seed} CSF={opts.csf_alloc}\n")
