"""Dense factor-matrix ops with gfx950 kernels where rocBLAS shapes are
degenerate (tall-skinny Gram). Reference: mat_aTa (src/matrix.c:414)."""
from __future__ import annotations

import torch

from splatt_amd._ext import native


def gram(A: torch.Tensor) -> torch.Tensor:
    """A^T A for row-major (n x F). Uses the HIP gram kernel on device for
    F <= 64 (rocBLAS picks a one-workgroup tile there); rocBLAS otherwise."""
    n, F = A.shape
    if A.device.type == "cuda" and F <= 64 and A.is_contiguous():
        G = torch.zeros(F, F, dtype=A.dtype, device=A.device)
        native().gpu_gram(A, G, torch.cuda.current_stream().cuda_stream)
        return G
    return A.T @ A
