"""CSF (Compressed Sparse Fiber) tensors, device-resident.

Capability parity: reference src/csf.{h,c} (csf_alloc policies csf.c:770-814,
mode orders csf.c:694-726, construction csf.c:248-644). Two builders:
  * CPU: the C++ core (parallel filter + prefix sums).
  * GPU: rocPRIM-backed torch primitives — stable radix sorts per level,
    boundary flags, cumsum compaction — the K13 "CSF construction on device"
    path (reference's hybrid counting sort, sort.c:761-905, reimagined as
    GPU radix sort + scans).
The structure is identical on both: flat fptr/fids per level + vals.
"""
from __future__ import annotations

from dataclasses import dataclass
from enum import Enum
from typing import List, Optional, Sequence

import torch

from splatt_amd._ext import native
from splatt_amd.sptensor import SpTensor


class CsfAllocPolicy(str, Enum):
    ONEMODE = "one"
    TWOMODE = "two"
    ALLMODE = "all"


@dataclass
class Csf:
    dims: List[int]                       # tensor dims (mode order)
    dim_perm: List[int]                   # level -> mode
    fptr: List[Optional[torch.Tensor]]    # per level int64; None at leaf
    fids: List[Optional[torch.Tensor]]    # per level int32; None root = dense
    vals: torch.Tensor

    @property
    def nmodes(self) -> int:
        return len(self.dims)

    @property
    def nnz(self) -> int:
        return int(self.vals.numel())

    @property
    def device(self) -> torch.device:
        return self.vals.device

    def level_of_mode(self, mode: int) -> int:
        return self.dim_perm.index(mode)

    def nfibs(self, level: int) -> int:
        if level == self.nmodes - 1:
            return self.nnz
        fp = self.fptr[level]
        return int(fp.numel() - 1) if fp is not None else 0

    def to(self, device) -> "Csf":
        c = Csf(
            list(self.dims), list(self.dim_perm),
            [t.to(device) if t is not None else None for t in self.fptr],
            [t.to(device) if t is not None else None for t in self.fids],
            self.vals.to(device),
        )
        # carry expansions + stage metadata (a frozen flat CSF has no fptr
        # tree to recompute them from)
        pack = getattr(self, "_pack", None)
        if pack is not None:
            p = pack.to(device)
            levels = list(self._pack_levels)  # type: ignore[attr-defined]
            object.__setattr__(c, "_pack", p)
            object.__setattr__(c, "_pack_levels", levels)
            object.__setattr__(c, "_expand_cache",
                               {l: p[:, col] for col, l in enumerate(levels)})
            c.fids[self.nmodes - 1] = c._expand_cache[self.nmodes - 1]  # type: ignore[attr-defined]
        else:
            cache = getattr(self, "_expand_cache", None)
            if cache:
                object.__setattr__(c, "_expand_cache",
                                   {l: t.to(device) for l, t in cache.items()})
        stage = getattr(self, "_stage", None)
        if stage is not None:
            object.__setattr__(c, "_stage", dict(stage))
        return c

    def storage_bytes(self) -> int:
        b = self.vals.numel() * self.vals.element_size()
        pack = getattr(self, "_pack", None)
        if pack is not None:
            # expansions/leaf fids are strided views into the pack
            return b + pack.numel() * pack.element_size()
        for t in self.fptr + self.fids:
            if t is not None:
                b += t.numel() * t.element_size()
        # flat builds keep contiguous expansions instead of the tree
        cache = getattr(self, "_expand_cache", None)
        if cache and all(fp is None for fp in self.fptr):
            for l, t in cache.items():
                if t is not self.fids[self.nmodes - 1]:
                    b += t.numel() * t.element_size()
        return b

    def ancestor_expand(self, level: int) -> torch.Tensor:
        """Per-nonzero ancestor label at `level` (int32, length nnz) — the
        sorted coordinate column recovered from the CSF compression. Feeds
        the flat MTTKRP kernels; cached after first use."""
        cache = getattr(self, "_expand_cache", None)
        if cache is None:
            cache = {}
            object.__setattr__(self, "_expand_cache", cache)
        if level in cache:
            return cache[level]
        nm = self.nmodes
        if level == nm - 1:
            t = self.fids[nm - 1]
        else:
            # nnz start of each node at `level`: compose fptr chains down
            nnzstart = self.fptr[nm - 2]
            for l in range(nm - 3, level - 1, -1):
                nnzstart = nnzstart[self.fptr[l]]
            counts = nnzstart[1:] - nnzstart[:-1]
            labels = self.fids[level]
            if labels is None:  # dense root: identity labels
                labels = torch.arange(counts.numel(), dtype=torch.int32,
                                      device=counts.device)
            t = torch.repeat_interleave(labels, counts)
        cache[level] = t
        return t

    def freeze_flat(self) -> "Csf":
        """Materialize every per-level expansion and drop the fptr tree —
        the flat MTTKRP kernels only need expansions + vals. Halves the
        device footprint of billion-nnz ALLMODE sets (fptr is int64 and
        ~nnz-long when fibers are short)."""
        for l in range(self.nmodes):
            self.ancestor_expand(l)
        self.fptr = [None] * self.nmodes
        return self

    def to_dict(self) -> dict:
        return {
            "fptr": [t.cpu() if t is not None else None for t in self.fptr],
            "fids": [t.cpu() if t is not None else None for t in self.fids],
            "vals": self.vals.cpu(),
            "dims": list(self.dims),
            "dim_perm": list(self.dim_perm),
            "nfibs": [self.nfibs(l) for l in range(self.nmodes)],
        }


def order_modes(dims: Sequence[int], policy: str, mode: int = 0) -> List[int]:
    """Mode-order policies: smallfirst / root / leaf."""
    return [int(x) for x in native().order_modes(list(dims), policy, mode)]


def build_csf(t: SpTensor, perm: Sequence[int], flat_only: bool = False,
              gather_tiles: int = 0, stage_rank: int = 0,
              lds_kb: int = 0) -> Csf:
    """Build one CSF with level->mode permutation `perm`. `flat_only`
    (device builds): skip the fptr/fids tree — the sorted columns ARE the
    flat kernel's expansions, so billion-nnz ALLMODE sets build ~2x faster
    at half the transient footprint. `gather_tiles` > 1 (flat_only builds):
    the dense-tiling analog for the flat kernel — nonzeros are bucketed by
    the row range of the LARGEST non-root mode (outermost sort key), so
    each phase's random factor-row gathers hit a 1/T-sized working set
    (reference tt_densetile's cache story, tile.c:262, recast for the
    per-XCD L2/L3 hierarchy). Output-key runs stay contiguous per bucket,
    so the kernel is unchanged (one atomic per key run per bucket)."""
    if any(d > 0xFFFFFFFF for d in t.dims):
        raise ValueError("mode dimensions above 2^32 are not supported "
                         "(CSF node ids are 32-bit; shard the mode first)")
    if t.device.type == "cuda":
        # device streams store labels as SIGNED int32 (the HIP kernels'
        # key/idx width); the host builder's unsigned ids reach 2^32
        if any(d > 0x7FFFFFFF for d in t.dims):
            raise ValueError(
                "device builds support mode dimensions up to 2^31-1 "
                "(int32 stream labels; shard the mode or build on CPU)")
        if t.nnz > 0x7FFFFFFF:
            raise ValueError(
                "device builds support up to 2^31-1 nonzeros per shard "
                "(torch sort limit; shard the tensor or build on CPU)")
        if lds_kb <= 0:
            import os
            lds_kb = int(os.environ.get("SPLATT_LDS_KB", "24"))
        return _build_csf_device(t, list(perm), flat_only, gather_tiles,
                                 stage_rank, lds_kb)
    d = native().csf_build(t.inds, t.vals, list(t.dims), list(perm))
    return Csf(dims=[int(x) for x in d["dims"]],
               dim_perm=[int(x) for x in d["dim_perm"]],
               fptr=[x if isinstance(x, torch.Tensor) else None for x in d["fptr"]],
               fids=[x if isinstance(x, torch.Tensor) else None for x in d["fids"]],
               vals=d["vals"])


def _build_csf_device(t: SpTensor, perm: List[int],
                      flat_only: bool = False,
                      gather_tiles: int = 0, stage_rank: int = 0,
                      lds_kb: int = 48) -> Csf:
    """All-device CSF construction with torch/rocPRIM primitives."""
    nm, nnz = t.nmodes, t.nnz
    dev = t.device
    if nnz == 0:
        c = Csf(dims=list(t.dims), dim_perm=list(perm),
                fptr=[torch.zeros(1, dtype=torch.int64, device=dev)
                      if l < nm - 1 else None for l in range(nm)],
                fids=[None if l == 0 else
                      torch.zeros(0, dtype=torch.int32, device=dev)
                      for l in range(nm)],
                vals=t.vals.clone())
        return c
    # lexicographic stable sort: least-significant level first
    order = torch.arange(nnz, device=dev)
    for level in reversed(range(nm)):
        keys = t.inds[perm[level]].index_select(0, order)
        order = order.index_select(0, torch.argsort(keys, stable=True))
    stage_meta = None
    cache_mb = 0
    if flat_only:
        import os as _os2
        cache_mb = int(_os2.environ.get("SPLATT_CACHE_TILE_MB", "0"))
    if flat_only and cache_mb > 0 and stage_rank > 0:
        # EXPERIMENTAL 3-D cache tiling for HBM-bound shapes: bucket by a
        # cell over ALL modes with per-mode factor windows <= cache_mb, so
        # every stream (output atomics + both gathers) works in an
        # L3-resident window. Output runs shrink to nnz/(ncells*root_dim);
        # the kernel is unchanged (atomic per run).
        vbytes = t.vals.element_size()
        win = cache_mb * 1024 * 1024
        cell = torch.zeros(nnz, dtype=torch.int64, device=dev)
        ncells = 1
        for l in range(nm):
            d = t.dims[perm[l]]
            tl = max(1, (d * stage_rank * vbytes + win - 1) // win)
            ch = (d + tl - 1) // tl
            b = torch.div(t.inds[perm[l]].index_select(0, order), ch,
                          rounding_mode="floor")
            cell = cell * tl + b
            ncells *= tl
        order = order.index_select(0, torch.argsort(cell, stable=True))
        stage_rank = 0  # no LDS staging on this sort order
    elif flat_only and (gather_tiles > 1 or stage_rank > 0):
        big = max(range(1, nm), key=lambda l: t.dims[perm[l]])
        if stage_rank > 0:
            # LDS-staging buckets: rows-per-bucket sized to the LDS budget.
            # Candidate selection: among non-root levels, stage the LARGEST
            # dim whose bucketing keeps the output-key runs dense — the
            # gate (>= min_run nnz per bucket x output row) protects the
            # one-atomic-per-run economy (measured: Netflix +13% at 8-9
            # nnz runs, -28% at ~1 nnz runs).
            import os as _os
            min_run = int(_os.environ.get("SPLATT_STAGE_MIN_RUN", "8"))
            # LDS tile sized by the factor STORAGE width (f32/bf16 store
            # modes fit 2-4x more rows per bucket)
            vbytes = {"f32": 4, "bf16": 2}.get(
                _os.environ.get("SPLATT_FACTOR_STORE", ""),
                t.vals.element_size())
            chunk_cap = max(64, (lds_kb * 1024) // (stage_rank * vbytes))
            root_dim = max(1, t.dims[perm[0]])
            # note: even L2-resident factors profit from staging (the TA
            # tag path, not bandwidth, is the binder — profiles/); the
            # size gate is available for experiments but off by default
            min_bytes = int(_os.environ.get("SPLATT_STAGE_MIN_BYTES", "0"))
            big, chunk, tiles = -1, 0, 1
            for l in sorted(range(1, nm), key=lambda x: -t.dims[perm[x]]):
                d = t.dims[perm[l]]
                if d * stage_rank * vbytes < min_bytes:
                    continue
                ch = min(chunk_cap, d)
                ti = (d + ch - 1) // ch
                if ti <= 1 or nnz // (ti * root_dim) >= min_run:
                    big, chunk, tiles = l, ch, ti
                    break
            if big < 0:
                stage_rank = 0
        else:
            dim_big = t.dims[perm[big]]
            tiles = gather_tiles
            chunk = (dim_big + tiles - 1) // tiles
        if tiles > 1:
            bucket = torch.div(t.inds[perm[big]].index_select(0, order),
                               chunk, rounding_mode="floor")
            order = order.index_select(0, torch.argsort(bucket, stable=True))
        if stage_rank > 0:
            stage_meta = {"level": big, "chunk": int(chunk),
                          "nbuckets": int(tiles)}
    sinds = [t.inds[perm[l]].index_select(0, order) for l in range(nm)]
    svals = t.vals.index_select(0, order)

    if flat_only:
        c = Csf(dims=list(t.dims), dim_perm=list(perm),
                fptr=[None] * nm, fids=[None] * nm, vals=svals)
        import os as _os3
        if (stage_meta is not None and nm <= 4
                and _os3.environ.get("SPLATT_PACK") != "0"):
            # packed stream: ONE int4 word per nonzero [key, staged level,
            # remaining levels] for the v6 LDS kernel (halves the stream
            # tag lookups, csrc/hip/mttkrp_lds.hip). The per-level
            # expansions become strided VIEWS into the pack — no second
            # copy of the index streams.
            lvl = stage_meta["level"]
            levels = [0, lvl] + [l for l in range(1, nm) if l != lvl]
            pack = torch.zeros(nnz, 4, dtype=torch.int32, device=dev)
            for col, l in enumerate(levels):
                pack[:, col] = sinds[l].to(torch.int32)
            cache = {l: pack[:, col] for col, l in enumerate(levels)}
            object.__setattr__(c, "_pack", pack)
            object.__setattr__(c, "_pack_levels", levels)
        else:
            cache = {l: sinds[l].to(torch.int32) for l in range(nm)}
        c.fids[nm - 1] = cache[nm - 1]
        object.__setattr__(c, "_expand_cache", cache)
        if stage_meta is not None:
            object.__setattr__(c, "_stage", stage_meta)
        return c

    # new-node flags per level: node at level l starts where any of levels
    # 0..l changes (diff-level trick, same invariant as the C++ builder)
    fptr: List[Optional[torch.Tensor]] = [None] * nm
    fids: List[Optional[torch.Tensor]] = [None] * nm
    change = torch.zeros(nnz, dtype=torch.bool, device=dev)
    change[0] = True
    starts_per_level: List[torch.Tensor] = []
    for l in range(nm - 1):
        if nnz > 1:
            change[1:] |= sinds[l][1:] != sinds[l][:-1]
        starts = change.nonzero(as_tuple=True)[0]
        starts_per_level.append(starts)
        fids[l] = sinds[l].index_select(0, starts).to(torch.int32)
    for l in range(nm - 1):
        starts = starts_per_level[l]
        if l < nm - 2:
            child = starts_per_level[l + 1]
            ptr = torch.searchsorted(child, starts)
            ptr = torch.cat([ptr, torch.tensor([child.numel()], device=dev)])
        else:
            ptr = torch.cat([starts, torch.tensor([nnz], device=dev)])
        fptr[l] = ptr.to(torch.int64)
    fids[nm - 1] = sinds[nm - 1].to(torch.int32)

    # dense root => identity labels dropped (reference NULL root fids)
    if fids[0] is not None and fids[0].numel() == t.dims[perm[0]]:
        fids[0] = None

    return Csf(dims=list(t.dims), dim_perm=list(perm),
               fptr=fptr, fids=fids, vals=svals)


@dataclass
class CsfSet:
    """One or more CSF copies + the per-output-mode dispatch map
    (reference csf_alloc / mode_csf_map, csf.c:770-814, mttkrp.c:1831-1861)."""
    csfs: List[Csf]
    mode_csf: List[int]
    mode_depth: List[int]

    @property
    def nmodes(self) -> int:
        return len(self.mode_csf)

    @property
    def dims(self) -> List[int]:
        return self.csfs[0].dims

    @property
    def nnz(self) -> int:
        return self.csfs[0].nnz

    def to(self, device) -> "CsfSet":
        return CsfSet([c.to(device) for c in self.csfs],
                      list(self.mode_csf), list(self.mode_depth))

    def storage_bytes(self) -> int:
        return sum(c.storage_bytes() for c in self.csfs)


def csf_alloc(t: SpTensor,
              policy: CsfAllocPolicy | str = CsfAllocPolicy.TWOMODE) -> CsfSet:
    policy = CsfAllocPolicy(policy)
    nm = t.nmodes
    if policy == CsfAllocPolicy.ONEMODE:
        perm = order_modes(t.dims, "smallfirst")
        c = build_csf(t, perm)
        return CsfSet([c], [0] * nm, [c.level_of_mode(m) for m in range(nm)])
    if policy == CsfAllocPolicy.TWOMODE:
        perm = order_modes(t.dims, "smallfirst")
        longest = perm[-1]
        c0 = build_csf(t, perm)
        c1 = build_csf(t, order_modes(t.dims, "root", longest))
        mode_csf, mode_depth = [], []
        for m in range(nm):
            if m == longest:
                mode_csf.append(1)
                mode_depth.append(0)
            else:
                mode_csf.append(0)
                mode_depth.append(c0.level_of_mode(m))
        return CsfSet([c0, c1], mode_csf, mode_depth)
    # ALLMODE
    csfs = [build_csf(t, order_modes(t.dims, "root", m)) for m in range(nm)]
    return CsfSet(csfs, list(range(nm)), [0] * nm)
