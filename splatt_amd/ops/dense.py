"""Dense factor-matrix ops with gfx950 kernels where rocBLAS shapes are
degenerate (tall-skinny Gram). Reference: mat_aTa (src/matrix.c:414)."""
from __future__ import annotations

import torch

from splatt_amd._ext import native


def gram(A: torch.Tensor) -> torch.Tensor:
    """A^T A for row-major (n x F). Uses the HIP gram kernel on device for
    F <= 64 (rocBLAS picks a one-workgroup tile there); rocBLAS otherwise.
    SPLATT_DETERMINISTIC=1: the serial-dot kernel (the MFMA gram lands
    per-wave partials with atomic adds, whose order varies run to run)."""
    import os
    n, F = A.shape
    if (os.environ.get("SPLATT_DETERMINISTIC") == "1"
            and A.device.type == "cuda" and F <= 64 and A.is_contiguous()):
        nparts = max(1, min(256, (n + 63) // 64))
        Gpart = torch.empty(nparts * F * F, dtype=A.dtype, device=A.device)
        G = torch.empty(F, F, dtype=A.dtype, device=A.device)
        native().gpu_gram_det(A, Gpart, G,
                              torch.cuda.current_stream().cuda_stream)
        return G
    if A.device.type == "cuda" and F <= 64 and A.is_contiguous():
        G = torch.zeros(F, F, dtype=A.dtype, device=A.device)
        native().gpu_gram(A, G, torch.cuda.current_stream().cuda_stream)
        return G
    return A.T @ A


def spd_inverse(G: torch.Tensor) -> torch.Tensor:
    """Inverse of an SPD F x F matrix. Device path (F <= 64): the
    one-workgroup HIP Cholesky kernel — hipGraph-capture-safe, no host
    sync, one launch instead of rocSOLVER's potrf/potri chain. Includes
    escalating Tikhonov jitter on breakdown (reference gelss fallback
    analog, matrix.c:554-599)."""
    F = G.shape[0]
    if G.device.type == "cuda" and F <= 64 and G.dtype in (torch.float64,
                                                           torch.float32):
        Ginv = torch.empty_like(G)
        native().gpu_spd_inverse(G.contiguous(), Ginv,
                                 torch.cuda.current_stream().cuda_stream)
        return Ginv
    eye = torch.eye(F, dtype=G.dtype, device=G.device)
    scale = float(G.diagonal().abs().max().clamp(min=1.0))
    jitter = 1e-12
    for _ in range(12):   # escalating Tikhonov (reference gelss fallback
        try:              # role, matrix.c:554-599)
            L = torch.linalg.cholesky(G + jitter * scale * eye)
            return torch.cholesky_inverse(L)
        except torch.linalg.LinAlgError:
            jitter *= 100.0
    # last resort: pseudo-inverse (rank-deficient Gram)
    return torch.linalg.pinv(G)


def solve_rows(mb: torch.Tensor, Ginv: torch.Tensor) -> torch.Tensor:
    """mb @ Ginv — the CPD solve GEMM. Default: library GEMM. With
    SPLATT_DETERMINISTIC=1 on device (F in the spec set): the one-thread-
    per-element HIP kernel (no split-K/atomics), so repeated runs are
    bitwise identical."""
    import os
    F = mb.shape[1]
    if (os.environ.get("SPLATT_DETERMINISTIC") == "1"
            and mb.device.type == "cuda" and F in (4, 8, 16, 32, 64)):
        C = torch.empty_like(mb)
        native().gpu_rowsolve(mb.contiguous(), Ginv.contiguous(), C,
                              torch.cuda.current_stream().cuda_stream)
        return C
    return mb @ Ginv
