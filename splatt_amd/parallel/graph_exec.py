"""hipGraph-captured ALS iteration (single GPU).

The steady-state ALS step is launch-bound on its dense tail (~50 small
kernels: Hadamard Gram chain, Cholesky + inverse, solve GEMM, normalize,
gram, fit) around 3 large MTTKRP launches. This runner rewrites the step
over STATIC buffers (every op is out=/in-place) and captures one full
iteration into a hipGraph (torch.cuda.CUDAGraph on ROCm); replay then
costs one launch. The fit is accumulated on device and only read back
after the timed region. Max-norm (it>=1) schedule only — iteration 0 runs
eager. Falls back to the eager step if capture fails.
"""
from __future__ import annotations

from typing import Optional

import torch

from splatt_amd.mttkrp import mttkrp
from splatt_amd._ext import native


class GraphStepRunner:
    """world==1, device-resident, static-shape ALS iteration."""

    def __init__(self, st):
        self.st = st
        self.graph: Optional[torch.cuda.CUDAGraph] = None
        dec = st.dec
        nm = len(dec.global_dims)
        F = st.factors[0].shape[1]
        dev = st.buf.device
        dtype = st.buf.dtype
        self.nm, self.F = nm, F
        # static buffers
        self.A = [torch.empty_like(st.factors[m]) for m in range(nm)]
        for m in range(nm):
            self.A[m].copy_(st.factors[m])
        self.G = torch.empty(F, F, dtype=dtype, device=dev)
        self.Ginv = torch.empty(F, F, dtype=dtype, device=dev)
        self.grams = [torch.empty(F, F, dtype=dtype, device=dev)
                      for _ in range(nm)]
        for m in range(nm):
            self.grams[m].copy_(st.grams[m])
        self.lam = torch.empty(F, dtype=dtype, device=dev)
        self.lam.copy_(st.lam)
        self.fit_parts = torch.zeros(2, dtype=torch.float64, device=dev)
        self.buf = st.buf
        self.tmp = torch.empty_like(st.buf)
        # deterministic mode: static workspaces for the atomic-free
        # solve/gram kernels (mttkrp already reads the env per call)
        import os
        self.det = (os.environ.get("SPLATT_DETERMINISTIC") == "1"
                    and F in (4, 8, 16, 32, 64))
        if self.det:
            self.gpart = []
            for m in range(nm):
                nparts = max(1, min(256, (dec.chunkn[m] + 63) // 64))
                self.gpart.append(torch.empty(nparts * F * F, dtype=dtype,
                                              device=dev))

    def _static_step(self):
        st, nm = self.st, self.nm
        dec = st.dec
        stream = torch.cuda.current_stream().cuda_stream
        for m in range(nm):
            n = dec.chunkn[m]
            mb = self.buf[:n]
            mttkrp(st.cs, self.A, m, out=mb)
            # G = hadamard of other grams (+jitter), L = chol, Ginv
            first = True
            for o in range(nm):
                if o == m:
                    continue
                if first:
                    self.G.copy_(self.grams[o])
                    first = False
                else:
                    self.G.mul_(self.grams[o])
            native().gpu_spd_inverse(self.G, self.Ginv, stream)
            if self.det:
                native().gpu_rowsolve(mb, self.Ginv, self.A[m], stream)
            else:
                torch.mm(mb, self.Ginv, out=self.A[m])
            # max-norm normalize (steady state)
            torch.amax(torch.abs(self.A[m]), dim=0, out=self.lam)
            self.lam.clamp_(min=1.0)
            self.A[m].div_(self.lam)
            if self.det:
                native().gpu_gram_det(self.A[m], self.gpart[m],
                                      self.grams[m], stream)
            else:
                self.grams[m].zero_()
                native().gpu_gram(self.A[m], self.grams[m], stream)
        # fit parts on device: [inner, knorm]
        mlast = nm - 1
        n = dec.chunkn[mlast]
        t = self.tmp[:n]
        torch.mul(self.buf[:n], self.A[mlast], out=t)
        inner_cols = t.sum(dim=0)          # small alloc (F) — capture-safe
        self.fit_parts[0] = (inner_cols.double() * self.lam.double()).sum()
        first = True
        for o in range(nm):
            if first:
                self.G.copy_(self.grams[o])
                first = False
            else:
                self.G.mul_(self.grams[o])
        lamd = self.lam.double()
        self.fit_parts[1] = (self.G.double()
                             * torch.outer(lamd, lamd)).sum()

    def capture(self) -> bool:
        # capture can fail transiently (allocator/stream state on a busy
        # device); one clean retry after a full sync before giving up
        for attempt in range(2):
            self.capture_error = None
            try:
                s = torch.cuda.Stream()
                with torch.cuda.stream(s):
                    for _ in range(2):   # warm up allocator on side stream
                        self._static_step()
                torch.cuda.current_stream().wait_stream(s)
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g):
                    self._static_step()
                self.graph = g
                return True
            except Exception as e:  # noqa: BLE001 - fall back to eager
                self.capture_error = repr(e)
                self.graph = None
                if attempt == 0:
                    torch.cuda.synchronize()
        return False

    def replay(self):
        self.graph.replay()

    def finalize(self, norm_x: float) -> float:
        """Push results back into the state; return the fit."""
        import math
        st = self.st
        for m in range(self.nm):
            st.factors[m] = self.A[m]
            st.grams[m] = self.grams[m]
        st.lam = self.lam
        parts = self.fit_parts.cpu()
        inner, knorm = float(parts[0]), float(parts[1])
        residual = math.sqrt(max(0.0, norm_x + knorm - 2 * inner))
        st.fit = 1.0 - residual / math.sqrt(norm_x)
        return st.fit
