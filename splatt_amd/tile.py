"""Dense tiling of nonzeros over an nmodes-dimensional tile grid.

Capability parity: reference src/tile.c — tt_densetile (:262 rearranges
nonzeros into a dense grid of tiles), tile-id math get_tile_id /
fill_tile_coords (tile.h:31-123) and the layer traversal get_next_tileid
(:398) that lets each worker own a disjoint layer of tiles.

In the MI355X engine the flat MTTKRP kernel's run-length fold already makes
output rows conflict-cheap, so tiling here serves (a) capability parity,
(b) cache-locality experiments (leaf-row-range tiling), and (c) the CPU
OpenMP path.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import List, Sequence

import torch

from splatt_amd.sptensor import SpTensor


@dataclass
class DenseTiling:
    tile_dims: List[int]          # tiles along each mode
    tile_ptr: torch.Tensor        # [ntiles+1] nnz ranges after reorder
    order: torch.Tensor           # nnz permutation applied

    @property
    def ntiles(self) -> int:
        return int(self.tile_ptr.numel() - 1)


def tile_id(coords: Sequence[int], tile_dims: Sequence[int]) -> int:
    """Row-major tile id from per-mode tile coordinates (tile.h:31)."""
    tid = 0
    for c, d in zip(coords, tile_dims):
        tid = tid * d + c
    return tid


def tile_coords(tid: int, tile_dims: Sequence[int]) -> List[int]:
    """Inverse of tile_id (fill_tile_coords, tile.h:66)."""
    out = [0] * len(tile_dims)
    for m in reversed(range(len(tile_dims))):
        out[m] = tid % tile_dims[m]
        tid //= tile_dims[m]
    return out


def next_tileid_in_layer(prev: int, tile_dims: Sequence[int], mode: int,
                         layer: int) -> int:
    """Iterate all tiles whose mode-`mode` coordinate equals `layer`
    (reference get_next_tileid, tile.c:398). Returns -1 when exhausted."""
    nm = len(tile_dims)
    if prev < 0:
        coords = [0] * nm
        coords[mode] = layer
        return tile_id(coords, tile_dims)
    coords = tile_coords(prev, tile_dims)
    for m in reversed(range(nm)):
        if m == mode:
            continue
        coords[m] += 1
        if coords[m] < tile_dims[m]:
            return tile_id(coords, tile_dims)
        coords[m] = 0
    return -1


def densetile(t: SpTensor, tile_dims: Sequence[int]) -> tuple[SpTensor, DenseTiling]:
    """Reorder nonzeros into row-major tile order; tiles split each mode's
    index range into `tile_dims[m]` near-equal chunks (tt_densetile,
    tile.c:262). Stable within a tile."""
    nm = t.nmodes
    tid = torch.zeros(t.nnz, dtype=torch.int64)
    for m in range(nm):
        chunk = (t.dims[m] + tile_dims[m] - 1) // tile_dims[m]
        tid = tid * tile_dims[m] + torch.div(t.inds[m], chunk,
                                             rounding_mode="floor")
    order = torch.argsort(tid, stable=True)
    ntiles = 1
    for d in tile_dims:
        ntiles *= d
    counts = torch.bincount(tid, minlength=ntiles)
    ptr = torch.zeros(ntiles + 1, dtype=torch.int64)
    ptr[1:] = torch.cumsum(counts, 0)
    out = SpTensor(t.inds[:, order], t.vals[order], list(t.dims))
    return out, DenseTiling(list(tile_dims), ptr, order)
