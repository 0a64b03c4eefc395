"""COO sparse tensor, torch-backed.

Capability parity: reference src/sptensor.{h,c} (struct sptensor.h:27-41,
dedup/empty-slice repair sptensor.c:135-229, stats stats.c:26-50) and the
I/O layer src/io.c (.tns text with 0/1-index autodetect, .bin binary).
The heavy lifting lives in the C++ core; this class is the Python face.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional, Sequence

import torch

from splatt_amd._ext import native


@dataclass
class SpTensor:
    inds: torch.Tensor            # [nmodes, nnz] int64
    vals: torch.Tensor            # [nnz] float32/float64
    dims: List[int]
    indmaps: List[Optional[torch.Tensor]] = field(default_factory=list)

    @property
    def nmodes(self) -> int:
        return int(self.inds.shape[0])

    @property
    def nnz(self) -> int:
        return int(self.inds.shape[1])

    @property
    def device(self) -> torch.device:
        return self.vals.device

    @property
    def dtype(self) -> torch.dtype:
        return self.vals.dtype

    def to(self, device) -> "SpTensor":
        return SpTensor(self.inds.to(device), self.vals.to(device),
                        list(self.dims), list(self.indmaps))

    def density(self) -> float:
        d = float(self.nnz)
        for s in self.dims:
            d /= s
        return d

    def normsq(self) -> float:
        return float(self.vals.double().square().sum())

    def unfold(self, mode: int) -> torch.Tensor:
        """Mode-`mode` matricization X_(m) as a sparse CSR matrix of shape
        (dims[mode], prod(other dims)); column index combines the remaining
        modes in increasing order (reference tt_unfold, sptensor.c:307-356,
        which produces the CSR unfolding consumed by mttkrp_giga)."""
        others = [m for m in range(self.nmodes) if m != mode]
        ncols = 1
        for m in others:
            ncols *= self.dims[m]
        col = torch.zeros(self.nnz, dtype=torch.int64, device=self.device)
        for m in others:
            col = col * self.dims[m] + self.inds[m]
        coo = torch.sparse_coo_tensor(
            torch.stack([self.inds[mode], col]), self.vals,
            (self.dims[mode], ncols)).coalesce()
        import warnings
        with warnings.catch_warnings():
            # torch's blanket "CSR support is in beta" notice
            warnings.simplefilter("ignore", UserWarning)
            return coo.to_sparse_csr()

    # ----------------------------------------------------------- repair ops

    def fixed(self, dedup: bool = True, compress: bool = False) -> "SpTensor":
        """Sort + merge duplicate nonzeros, optionally drop empty slices
        (the `splatt check` repairs, reference cmd_check.c:63-122)."""
        i, v, d, ndups, nempty, indmaps = native().coo_fix(
            self.inds.cpu(), self.vals.cpu(), list(self.dims), dedup, compress)
        maps = [m if isinstance(m, torch.Tensor) else None for m in indmaps]
        out = SpTensor(i.to(self.device), v.to(self.device), list(d), maps)
        out._ndups = int(ndups)     # type: ignore[attr-defined]
        out._nempty = int(nempty)   # type: ignore[attr-defined]
        return out

    # ------------------------------------------------------------------ io

    @staticmethod
    def load(path: str, dtype: torch.dtype = torch.float64) -> "SpTensor":
        tag = "f32" if dtype == torch.float32 else "f64"
        i, v, d = native().tensor_load(str(path), tag)
        return SpTensor(i, v, list(d))

    def save(self, path: str) -> None:
        path = str(path)
        if path.endswith(".bin"):
            native().bin_write(path, self.inds.cpu(), self.vals.cpu(),
                               list(self.dims), 8, self.vals.element_size())
        else:
            native().tns_write(path, self.inds.cpu(), self.vals.cpu(),
                               list(self.dims))

    # ------------------------------------------------------------ synthetic

    @staticmethod
    def synthetic(dims: Sequence[int], nnz: int,
                  dtype: torch.dtype = torch.float64,
                  device: str | torch.device = "cpu",
                  seed: int = 0x5eed,
                  concentration: float = 1.0,
                  dist: str = "uniform",
                  zipf_a: float = 1.1) -> "SpTensor":
        """Random synthetic tensor of a target shape/nnz.

        `dist`: "uniform" draws each index independently (hardest case
        for CSF compression — near-singleton fibers); "zipf" draws each
        mode's indices from a power law with exponent `zipf_a`, giving
        the heavy slices / long fibers real tensors show (the
        load-balance regime the reference's CCP + privatization target).
        `concentration` > 1 additionally skews uniform draws toward low
        ids. Duplicates are NOT merged (matches how nnz counts are
        quoted).
        """
        g = torch.Generator(device="cpu").manual_seed(seed)
        cols = []
        for m, d in enumerate(dims):
            u = torch.rand(nnz, generator=g, dtype=torch.float64)
            if dist == "zipf":
                # inverse-CDF of a bounded Pareto on [1, d]
                a = zipf_a
                lo, hi = 1.0, float(d)
                x = (hi ** (1 - a) - lo ** (1 - a)) * u + lo ** (1 - a)
                idx = x.pow(1.0 / (1 - a)).long() - 1
                # random relabeling so heavy slices are spread over ids
                perm = torch.randperm(d, generator=g)
                cols.append(perm[idx.clamp_(0, d - 1)])
            else:
                if concentration != 1.0:
                    u = u.pow(concentration)
                cols.append((u * d).long().clamp_(0, d - 1))
        inds = torch.stack(cols, 0)
        vals = torch.rand(nnz, generator=g, dtype=torch.float64).to(dtype)
        t = SpTensor(inds, vals, list(dims))
        return t.to(device)

    def stats(self) -> dict:
        bytes_coo = self.nnz * (8 * self.nmodes + self.vals.element_size())
        return {
            "nmodes": self.nmodes,
            "dims": list(self.dims),
            "nnz": self.nnz,
            "density": self.density(),
            "coo_bytes": bytes_coo,
        }
