"""Tensor -> graph / hypergraph conversions for partitioning-driven
reorderings.

Capability parity: reference src/graph.c — nonzero-vertex hypergraph
(hgraph_nnz_alloc:452), fiber-vertex hypergraph (hgraph_fib_alloc:506),
merged m-partite graph with shared-nnz edge weights (graph_convert:637),
uncut-hyperedge extraction (hgraph_uncut:576) and the METIS/PaToH-style
file writers (io.c:782-845). External partitioners (METIS/PaToH) are not
bundled — like the reference, we emit their input formats and consume
their output partition files.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import List

import torch

from splatt_amd.sptensor import SpTensor


@dataclass
class Hypergraph:
    nvtxs: int
    eptr: torch.Tensor   # [nhedges+1] int64
    eind: torch.Tensor   # pins, int64
    vwts: torch.Tensor | None = None

    @property
    def nhedges(self) -> int:
        return int(self.eptr.numel() - 1)


@dataclass
class Graph:
    nvtxs: int
    adj_ptr: torch.Tensor   # CSR [nvtxs+1]
    adj_ind: torch.Tensor
    adj_wgt: torch.Tensor
    vtx_dist: List[int] | None = None  # mode -> first vertex id


def hgraph_nnz(t: SpTensor) -> Hypergraph:
    """Vertices = nonzeros; one hyperedge per (mode, slice) containing the
    nonzeros of that slice (reference hgraph_nnz_alloc, graph.c:452)."""
    eptr = [torch.tensor([0], dtype=torch.int64)]
    eind = []
    for m in range(t.nmodes):
        order = torch.argsort(t.inds[m], stable=True)
        counts = torch.bincount(t.inds[m], minlength=t.dims[m])
        eptr.append(torch.cumsum(counts, 0) + (eptr[-1][-1] if eptr else 0))
        eind.append(order)
    ptr = torch.cat(eptr)
    return Hypergraph(nvtxs=t.nnz, eptr=ptr, eind=torch.cat(eind))


def hgraph_fib(t: SpTensor, mode: int) -> Hypergraph:
    """Vertices = mode-`mode` fibers (unique index tuples over the other
    modes... represented by grouping nnz); hyperedges = slices of every
    mode touched by each fiber (reference hgraph_fib_alloc, graph.c:506).
    Fibers here follow the smallest-first CSF sort with `mode` at the leaf.
    """
    from splatt_amd.csf import build_csf, order_modes
    perm = order_modes(t.dims, "leaf", mode)
    c = build_csf(t.to("cpu"), perm)
    nm = t.nmodes
    nfib = c.nfibs(nm - 2)
    # pins: for each (mode', slice) hyperedge, which fibers touch it
    fiber_of_nnz = torch.repeat_interleave(
        torch.arange(nfib, dtype=torch.int64),
        c.fptr[nm - 2][1:] - c.fptr[nm - 2][:-1])
    eptr = [0]
    eind = []
    for lv in range(nm):
        lab = c.ancestor_expand(lv).long()
        key = lab * nfib + fiber_of_nnz
        uniq = torch.unique(key)
        slice_id = uniq // nfib
        fib_id = uniq % nfib
        counts = torch.bincount(slice_id, minlength=c.dims[c.dim_perm[lv]])
        for cnt in counts.tolist():
            eptr.append(eptr[-1] + cnt)
        eind.append(fib_id[torch.argsort(slice_id, stable=True)])
    return Hypergraph(nvtxs=nfib, eptr=torch.tensor(eptr, dtype=torch.int64),
                      eind=torch.cat(eind))


def hgraph_uncut(hg: Hypergraph, part: torch.Tensor) -> torch.Tensor:
    """Ids of hyperedges whose pins all share one partition
    (reference hgraph_uncut, graph.c:576)."""
    out = []
    for e in range(hg.nhedges):
        pins = hg.eind[hg.eptr[e]: hg.eptr[e + 1]]
        if pins.numel() and (part[pins] == part[pins[0]]).all():
            out.append(e)
    return torch.tensor(out, dtype=torch.int64)


def graph_mpartite(t: SpTensor) -> Graph:
    """Merged m-partite graph: vertices = all slices of all modes; an edge
    (slice_a, slice_b) with weight = #shared nonzeros for every pair of
    modes (reference graph_convert, graph.c:637)."""
    nm = t.nmodes
    vtx_dist = [0]
    for m in range(nm):
        vtx_dist.append(vtx_dist[-1] + t.dims[m])
    nv = vtx_dist[-1]
    # accumulate edge multiset over all mode pairs
    keys = []
    for a in range(nm):
        for b in range(a + 1, nm):
            va = t.inds[a] + vtx_dist[a]
            vb = t.inds[b] + vtx_dist[b]
            keys.append(va * nv + vb)
            keys.append(vb * nv + va)
    allk = torch.cat(keys)
    uniq, cnt = torch.unique(allk, return_counts=True)
    src = (uniq // nv).long()
    dst = (uniq % nv).long()
    order = torch.argsort(src, stable=True)
    src, dst, cnt = src[order], dst[order], cnt[order]
    ptr = torch.zeros(nv + 1, dtype=torch.int64)
    ptr[1:] = torch.cumsum(torch.bincount(src, minlength=nv), 0)
    return Graph(nvtxs=nv, adj_ptr=ptr, adj_ind=dst, adj_wgt=cnt,
                 vtx_dist=vtx_dist)


def graph_write(g: Graph, path: str) -> None:
    """METIS graph format with edge weights (fmt 001)."""
    with open(path, "w") as f:
        f.write(f"{g.nvtxs} {g.adj_ind.numel() // 2} 001\n")
        for v in range(g.nvtxs):
            lo, hi = int(g.adj_ptr[v]), int(g.adj_ptr[v + 1])
            toks = []
            for i in range(lo, hi):
                toks.append(f"{int(g.adj_ind[i]) + 1} {int(g.adj_wgt[i])}")
            f.write(" ".join(toks) + "\n")


def hgraph_write(hg: Hypergraph, path: str) -> None:
    """PaToH-style hypergraph file (reference hgraph_write, io.c:782)."""
    with open(path, "w") as f:
        f.write(f"{hg.nhedges} {hg.nvtxs} {hg.eind.numel()}\n")
        for e in range(hg.nhedges):
            pins = hg.eind[hg.eptr[e]: hg.eptr[e + 1]]
            f.write(" ".join(str(int(p) + 1) for p in pins) + "\n")


def part_read(path: str) -> torch.Tensor:
    """Partition file: one integer per vertex (reference part_read)."""
    with open(path) as f:
        vals = [int(line.split()[0]) for line in f if line.strip()]
    return torch.tensor(vals, dtype=torch.int64)
