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


def stats_hparts(t, part: "torch.Tensor", name: str = "") -> str:
    """Partition-quality report for an nnz partition (fine-grained
    decomposition input): per-part nonzero balance plus, per mode, the
    number of cut slices and the average unique slice indices touched
    per part — the communication-volume proxy (reference p_stats_hparts,
    stats.c:53-169, recast on the nnz-vertex hypergraph that our
    fine-grained RCCL decomposition consumes)."""
    import torch
    nparts = int(part.max()) + 1
    sizes = torch.bincount(part, minlength=nparts)
    lines = ["Partition information ------------------------------------",
             f"FILE={name}" if name else "",
             f"NVTXS={t.nnz} NPARTS={nparts} "
             f"LIGHTEST={int(sizes.min())} HEAVIEST={int(sizes.max())} "
             f"AVG={t.nnz / nparts:.1f}"]
    for m in range(t.nmodes):
        rows = t.inds[m]
        # parts touching each slice
        key = rows * nparts + part
        touched = torch.unique(key)
        slices_touched = torch.unique(touched // nparts).numel()
        cut = touched.numel() - slices_touched       # extra (row, part) pairs
        uniq_per_part = touched.numel() / nparts
        lines.append(f"mode {m + 1}: cut slices (extra owners) = {cut}; "
                     f"avg unique rows/part = {uniq_per_part:.1f} "
                     f"of {t.dims[m]}")
    return "\n".join(x for x in lines if x)
