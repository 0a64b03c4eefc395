"""MTTKRP dispatch: gfx950 HIP kernels on device, C++ core on host.

Capability parity: reference src/mttkrp.c (`mttkrp_csf` dispatch :1287-1341,
`splatt_mttkrp` API :1763, `mttkrp_stream` gold oracle :1697-1757).
Device path: the CDNA4 kernels in csrc/hip/mttkrp_kernels.hip — the HIP
extension is REQUIRED on CUDA tensors (no eager fallback).
"""
from __future__ import annotations

import os
from typing import List, Optional

import torch

from splatt_amd._ext import native
from splatt_amd.csf import Csf, CsfSet
from splatt_amd.sptensor import SpTensor


def mttkrp_stream(t: SpTensor, mats: List[torch.Tensor], mode: int) -> torch.Tensor:
    """COO streaming gold oracle (CPU, deterministic)."""
    return native().mttkrp_stream(t.inds.cpu(), t.vals.cpu(), list(t.dims),
                                  [m.cpu() for m in mats], mode)


def _gpu_mttkrp_csf(c: Csf, depth: int, mats: List[torch.Tensor],
                    mode: int, out: torch.Tensor) -> None:
    nm = c.nmodes
    if nm != 3:
        raise NotImplementedError(
            f"the hierarchical walk kernels support 3-mode CSF, got {nm} "
            f"modes (use alg='flat')")
    if c.fptr[0] is None:
        raise ValueError(
            "this CSF was built flat-only (no fptr tree); the hierarchical "
            "walk kernels need a full build — use alg='flat'")
    perm = c.dim_perm
    which = 0 if depth == 0 else (2 if depth == nm - 1 else 1)
    if which == 0:
        ma, mb = mats[perm[1]], mats[perm[2]]
    elif which == 1:
        ma, mb = mats[perm[0]], mats[perm[2]]
    else:
        ma, mb = mats[perm[0]], mats[perm[1]]
    stream = torch.cuda.current_stream().cuda_stream
    native().gpu_mttkrp3(
        which, c.fptr[0], c.fids[0], c.fptr[1], c.fids[1], c.fids[2],
        c.vals, ma.contiguous(), mb.contiguous(), out, stream)


def _stage_blocks(c: Csf, rows: tuple | None = None) -> dict:
    """Per-workgroup (nnz range, bucket row0) descriptors for the
    LDS-staged kernel; cached on the Csf. `rows=(lo,hi)` restricts the
    descriptors to stream positions whose OUTPUT key (root label) lies in
    [lo, hi) — the stream is bucket-major but root-key-sorted inside each
    bucket, so the restriction is one searchsorted per bucket (done once,
    cached; enables the chunked comm/compute pipeline in parallel/grid.py)."""
    cache = getattr(c, "_stage_block_cache", None)
    if cache is None:
        cache = {}
        object.__setattr__(c, "_stage_block_cache", cache)
    ck = rows
    if ck in cache:
        return cache[ck]
    st = c._stage  # type: ignore[attr-defined]
    lvl, chunk, tiles = st["level"], st["chunk"], st["nbuckets"]
    dev = c.device
    nnz = c.nnz
    if tiles > 1:
        bucket = torch.div(c.ancestor_expand(lvl), chunk,
                           rounding_mode="floor").contiguous()
        bnd = torch.searchsorted(
            bucket, torch.arange(tiles + 1, dtype=torch.int32, device=dev),
            right=False).cpu()
    else:
        bnd = torch.tensor([0, nnz])
    segs = []          # (start, end, bucket_id) stream segments to cover
    if rows is None:
        for b in range(len(bnd) - 1):
            segs.append((int(bnd[b]), int(bnd[b + 1]), b))
    else:
        lo, hi = rows
        key = c.ancestor_expand(0)
        kb = torch.tensor([lo, hi], dtype=key.dtype, device=dev)
        for b in range(len(bnd) - 1):
            s, e = int(bnd[b]), int(bnd[b + 1])
            if s == e:
                continue
            pos = torch.searchsorted(key[s:e].contiguous(), kb).cpu()
            p0, p1 = s + int(pos[0]), s + int(pos[1])
            if p0 < p1:
                segs.append((p0, p1, b))
    tgt = max(4096, nnz // int(os.environ.get("SPLATT_LDS_BLOCKS", "16384")))
    starts, ends, row0s, bkt0s = [], [], [], []
    for s, e, b in segs:
        p = s
        while p < e:
            q = min(e, p + tgt)
            starts.append(p)
            ends.append(q)
            row0s.append(b * chunk)
            bkt0s.append(s)       # bucket's first stream position
            p = q
    blocks = {
        "start": torch.tensor(starts, dtype=torch.int64, device=dev),
        "end": torch.tensor(ends, dtype=torch.int64, device=dev),
        "row0": torch.tensor(row0s, dtype=torch.int32, device=dev),
        "bucket_p0": torch.tensor(bkt0s, dtype=torch.int64, device=dev),
        "chunk": chunk,
        "level": lvl,
    }
    cache[ck] = blocks
    return blocks


def _key_sorted(c: Csf, depth: int) -> bool:
    """Is the per-nnz output-key expansion non-decreasing? (Cached; true
    for root-output streams of unbucketed builds.)"""
    cache = getattr(c, "_key_sorted_cache", None)
    if cache is None:
        cache = {}
        object.__setattr__(c, "_key_sorted_cache", cache)
    if depth not in cache:
        key = c.ancestor_expand(depth)
        cache[depth] = bool((key[1:] >= key[:-1]).all()) if key.numel() > 1 \
            else True
    return cache[depth]


def _key_bounds(c: Csf, depth: int, lo: int, hi: int) -> tuple:
    """Stream positions [p0,p1) whose sorted output key is in [lo,hi)
    (cached ints — the one-time device sync happens on first use)."""
    cache = getattr(c, "_key_bounds_cache", None)
    if cache is None:
        cache = {}
        object.__setattr__(c, "_key_bounds_cache", cache)
    k = (depth, lo, hi)
    if k not in cache:
        key = c.ancestor_expand(depth).contiguous()
        kb = torch.tensor([lo, hi], dtype=key.dtype, device=key.device)
        pos = torch.searchsorted(key, kb).cpu()
        cache[k] = (int(pos[0]), int(pos[1]))
    return cache[k]


def _gpu_mttkrp_det(c: Csf, depth: int, mats: List[torch.Tensor],
                    out: torch.Tensor) -> None:
    """Bitwise-deterministic device MTTKRP (csrc/hip/mttkrp_det.hip):
    plain stores for walker-interior output runs + a walker-ordered fixup
    of span-boundary partials. Needs a depth-0 (root-sorted) stream —
    every key contiguous — i.e. the default ALLMODE policy; and a spec
    rank. The reference's reproducibility answer is its serial
    mttkrp_stream (src/mttkrp.c:1697); this keeps the property on-GPU."""
    nm = c.nmodes
    rank = int(mats[0].shape[1])
    if depth != 0:
        raise ValueError(
            "deterministic MTTKRP needs a root-sorted (depth-0) stream for "
            f"this mode; got depth {depth} — build with csf_alloc='all'")
    if rank not in (4, 8, 16, 32, 64) or nm > 5:
        raise ValueError(
            f"deterministic MTTKRP supports ranks 4/8/16/32/64 and <=5 "
            f"modes, got rank {rank}, {nm} modes")
    # LDS-staged deterministic kernel (det6): same packed bucket-major
    # stream as the default v6 path, output privatized per bucket, ordered
    # fixup + ascending-bucket fold (csrc/hip/mttkrp_det.hip). Used when
    # the bucket-partial workspace fits SPLATT_DET_MB (default 4096 MB).
    pack = getattr(c, "_pack", None)
    st6 = getattr(c, "_stage", None)
    if (pack is not None and st6 is not None and nm <= 4
            and os.environ.get("SPLATT_NO_DET6") != "1"):
        blocks = _stage_blocks(c)
        nbuckets = st6["nbuckets"]
        nrows = c.dims[c.dim_perm[0]]
        budget = int(os.environ.get("SPLATT_DET_MB", "4096")) << 20
        need_outb = nbuckets * nrows * rank * c.vals.element_size()
        if need_outb <= budget:
            nblocks = int(blocks["start"].numel())
            nwalk = nblocks * 4 * (64 // rank)
            ws = getattr(c, "_det6_ws", None)
            if (ws is None or ws[0].numel() < nbuckets * nrows * rank
                    or ws[1].numel() < nwalk * 2 * rank):
                outb = torch.empty(nbuckets * nrows * rank,
                                   dtype=c.vals.dtype, device=c.device)
                side = torch.empty(nwalk * 2 * rank, dtype=c.vals.dtype,
                                   device=c.device)
                ws = (outb, side)
                object.__setattr__(c, "_det6_ws", ws)
            levels = c._pack_levels  # type: ignore[attr-defined]
            ms = [mats[c.dim_perm[l]].contiguous() for l in levels[1:]]
            stream = torch.cuda.current_stream().cuda_stream
            native().gpu_mttkrp_det6(
                pack, ms, c.vals, blocks["start"], blocks["end"],
                blocks["row0"], blocks["bucket_p0"], blocks["chunk"],
                c.dims[c.dim_perm[st6["level"]]],   # staged level's dim
                nbuckets, ws[0], ws[1], out, stream)
            return
    # the deterministic scheme needs every output key CONTIGUOUS in the
    # stream. LDS-bucketed builds reorder the stream bucket-major, so a
    # key-sorted copy is built once and cached (stable argsort -> the
    # copy itself is deterministic).
    streams = getattr(c, "_det_streams", None)
    if streams is None:
        streams = {}
        object.__setattr__(c, "_det_streams", streams)
    if depth not in streams:
        key = c.ancestor_expand(depth)
        idx = [c.ancestor_expand(l) for l in range(nm) if l != depth]
        vals = c.vals
        if not bool((key[1:] >= key[:-1]).all()):
            perm = torch.argsort(key, stable=True)
            key = key.index_select(0, perm).contiguous()
            idx = [i.index_select(0, perm).contiguous() for i in idx]
            vals = vals.index_select(0, perm).contiguous()
        else:
            # packed builds expose strided views; the det kernel wants
            # contiguous streams
            key = key.contiguous()
            idx = [i.contiguous() for i in idx]
        streams[depth] = (key, idx, vals)
    key, idx, vals = streams[depth]
    ms = [mats[c.dim_perm[l]].contiguous() for l in range(nm) if l != depth]
    need = native().flat_det_ws_elems(c.nnz, rank)
    ws = getattr(c, "_det_ws", None)
    if (ws is None or ws.numel() < need or ws.dtype != c.vals.dtype
            or ws.device != c.device):
        ws = torch.empty(need, dtype=c.vals.dtype, device=c.device)
        object.__setattr__(c, "_det_ws", ws)
    stream = torch.cuda.current_stream().cuda_stream
    native().gpu_mttkrp_flat_det(key, idx, ms, vals, out, ws, stream)


def _use_lds(c: Csf, depth: int, rank: int) -> bool:
    return (depth == 0 and c.nmodes <= 5
            and getattr(c, "_stage", None) is not None
            and rank in (4, 8, 16, 32, 64)
            and os.environ.get("SPLATT_NO_LDS") != "1")


def _gpu_mttkrp_flat(c: Csf, depth: int, mats: List[torch.Tensor],
                     out: torch.Tensor, rows: tuple | None = None) -> None:
    """Flat expanded-CSF kernel (see csrc/hip/mttkrp_flat.hip): per-nnz
    product of the non-output modes' rows folded by runs of the output
    key. Root-output dispatch on bucketed builds uses the LDS-staged
    kernel (csrc/hip/mttkrp_lds.hip) when the rank is in the spec set.
    `rows=(lo,hi)` restricts the launch to output rows in [lo,hi)."""
    nm = c.nmodes
    key = c.ancestor_expand(depth)
    rank = int(mats[0].shape[1])
    stream = torch.cuda.current_stream().cuda_stream
    # reduced-precision STORAGE mats need the packed v6 or plain v2
    # kernels; staged-but-unpacked dispatches (5-mode builds, or
    # SPLATT_NO_PACK debugging) route to v2
    storage = mats[0].dtype != c.vals.dtype
    packed = (getattr(c, "_pack", None) is not None
              and os.environ.get("SPLATT_NO_PACK") != "1")
    if _use_lds(c, depth, rank) and (packed or not storage):
        blocks = _stage_blocks(c, rows)
        lvl = blocks["level"]
        if int(blocks["start"].numel()) == 0:
            return
        pack = getattr(c, "_pack", None)
        if pack is not None and os.environ.get("SPLATT_NO_PACK") != "1":
            # packed-stream v6: word order [key, staged, rest] fixed at
            # build time (csf.py); mats follow _pack_levels[1:]
            levels = c._pack_levels  # type: ignore[attr-defined]
            ms = [mats[c.dim_perm[l]].contiguous() for l in levels[1:]]
            native().gpu_mttkrp_flat6(
                pack, ms, c.vals, blocks["start"], blocks["end"],
                blocks["row0"], blocks["chunk"], c.dims[c.dim_perm[lvl]],
                out, stream)
            return
        idx = [c.ancestor_expand(lvl).contiguous()]
        ms = [mats[c.dim_perm[lvl]].contiguous()]
        for l in range(nm):
            if l in (depth, lvl):
                continue
            idx.append(c.ancestor_expand(l).contiguous())
            ms.append(mats[c.dim_perm[l]].contiguous())
        native().gpu_mttkrp_flat5(
            key.contiguous(), idx, ms, c.vals, blocks["start"],
            blocks["end"], blocks["row0"], blocks["chunk"],
            c.dims[c.dim_perm[lvl]], out, stream)
        return
    p0, p1 = 0, c.nnz
    if rows is not None:
        if not _key_sorted(c, depth):
            raise ValueError("rows-restricted MTTKRP needs a key-sorted "
                             "stream for this mode")
        p0, p1 = _key_bounds(c, depth, rows[0], rows[1])
        if p0 >= p1:
            return
    idx, ms = [], []
    for l in range(nm):
        if l == depth:
            continue
        idx.append(c.ancestor_expand(l)[p0:p1].contiguous())
        ms.append(mats[c.dim_perm[l]].contiguous())
    native().gpu_mttkrp_flat(key[p0:p1].contiguous(), idx, ms,
                             c.vals[p0:p1], out, stream)


def factor_store_dtype():
    """Optional reduced-precision factor STORAGE for the MTTKRP gathers
    (SPLATT_FACTOR_STORE=f32|bf16; accumulation stays in the tensor's
    dtype, f64). Cuts gathered cache lines 2-4x on HBM-bound shapes
    (documented mode — the default and every headline number stay f64)."""
    env = os.environ.get("SPLATT_FACTOR_STORE", "")
    return {"f32": torch.float32, "bf16": torch.bfloat16}.get(env)


def mttkrp_rows_ok(src: CsfSet | Csf, mode: int, rank: int) -> bool:
    """Can mttkrp() honor a rows=(lo,hi) restriction for this mode?
    True when the dispatch lands on the LDS-staged kernel or on a
    key-sorted stream (any root-output ALLMODE build). The chunked
    comm/compute pipeline (parallel/grid.py) probes this and falls back
    to the unchunked schedule otherwise."""
    if isinstance(src, CsfSet):
        c = src.csfs[src.mode_csf[mode]]
        depth = src.mode_depth[mode]
    else:
        c, depth = src, src.level_of_mode(mode)
    if os.environ.get("SPLATT_DETERMINISTIC") == "1":
        return False
    if c.device.type == "cuda" and _use_lds(c, depth, rank):
        return True
    return _key_sorted(c, depth)


def mttkrp(src: CsfSet | Csf, mats: List[torch.Tensor], mode: int,
           out: Optional[torch.Tensor] = None,
           nthreads: int = 0, alg: str = "flat",
           deterministic: Optional[bool] = None,
           rows: Optional[tuple] = None) -> torch.Tensor:
    """MTTKRP for output `mode`; `mats` indexed by tensor mode.

    Device algorithms: 'flat' (default, expanded-CSF streaming kernel) or
    'csf' (hierarchical fiber-walk kernels, 3-mode only) — the reference
    keeps multiple MTTKRP algorithms selectable the same way (bench.c).
    `deterministic` (or SPLATT_DETERMINISTIC=1) selects the
    bitwise-reproducible device kernel (no atomics; ALLMODE + spec ranks).
    `rows=(lo,hi)`: compute ONLY output rows in [lo,hi) (zeroing just that
    slice of `out`) — the building block of the chunked reduce-scatter
    pipeline; probe support with mttkrp_rows_ok().
    """
    if isinstance(src, CsfSet):
        c = src.csfs[src.mode_csf[mode]]
        depth = src.mode_depth[mode]
    else:
        c = src
        depth = c.level_of_mode(mode)

    if not 0 <= mode < c.nmodes:
        raise IndexError(f"mode {mode} out of range for {c.nmodes} modes")
    if len(mats) != c.nmodes:
        raise ValueError(f"need {c.nmodes} factor matrices, got {len(mats)}")
    rank = int(mats[0].shape[1])
    storage = (c.device.type == "cuda" and c.vals.dtype == torch.float64
               and mats[0].dtype in (torch.float32, torch.bfloat16))
    for m, A in enumerate(mats):
        if A.shape != (c.dims[m], rank):
            raise ValueError(
                f"factor {m} has shape {tuple(A.shape)}, expected "
                f"({c.dims[m]}, {rank})")
        if (A.dtype != c.vals.dtype and not storage) or A.device != c.device:
            raise ValueError(
                f"factor {m}: dtype/device {A.dtype}/{A.device} does not "
                f"match tensor {c.vals.dtype}/{c.device}")
    if storage and rank not in (4, 8, 16, 32, 64):
        raise ValueError(
            "reduced-precision factor storage (SPLATT_FACTOR_STORE) needs "
            f"rank in {{4,8,16,32,64}}, got {rank}")
    if out is None:
        out = torch.empty(c.dims[mode], rank, dtype=c.vals.dtype,
                          device=mats[0].device)
    if c.device.type == "cuda":
        if native().hip_arch() != 950:
            raise RuntimeError("HIP kernels not built for gfx950")
        if deterministic is None:
            deterministic = os.environ.get("SPLATT_DETERMINISTIC") == "1"
        if rows is not None:
            if deterministic or alg != "flat":
                raise ValueError("rows-restricted MTTKRP supports the "
                                 "default flat kernels only")
            out[rows[0]: rows[1]].zero_()
            _gpu_mttkrp_flat(c, depth, mats, out, rows)
            return out
        out.zero_()
        if deterministic:
            _gpu_mttkrp_det(c, depth, mats, out)
        elif alg == "flat":
            _gpu_mttkrp_flat(c, depth, mats, out)
        else:
            _gpu_mttkrp_csf(c, depth, mats, mode, out)
        return out
    if rows is not None:
        # host rows-restricted path (multi-process gloo tests): same math
        # via the sorted stream expansions + index_add_
        lo, hi = rows
        if not _key_sorted(c, depth):
            raise ValueError("rows-restricted MTTKRP needs a key-sorted "
                             "stream for this mode")
        p0, p1 = _key_bounds(c, depth, lo, hi)
        out[lo:hi].zero_()
        if p0 < p1:
            key = c.ancestor_expand(depth)[p0:p1].long()
            x = None
            for l in range(c.nmodes):
                if l == depth:
                    continue
                il = c.ancestor_expand(l)[p0:p1].long()
                r = mats[c.dim_perm[l]].index_select(0, il)
                x = r.clone() if x is None else x.mul_(r)
            x.mul_(c.vals[p0:p1].unsqueeze(1))
            out.index_add_(0, key, x)
        return out
    res = native().mttkrp_csf_cpu(c.to_dict(), [m.cpu() for m in mats],
                                  mode, nthreads)
    out.copy_(res)
    return out
