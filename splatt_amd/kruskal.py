"""Kruskal-tensor utilities.

Capability parity: the reference exposes norm/innerprod/dim over Kruskal
tensors through its Octave/Matlab layer (matlab/splatt_norm.m,
splatt_innerprod.m over the C API) and uses the same quantities inside
the fit computation (cpd.c:116-265). Here they are first-class Python
API over the factor matrices.
"""
from __future__ import annotations

import torch

from splatt_amd.cpd import Kruskal
from splatt_amd.sptensor import SpTensor


def kruskal_norm(k: Kruskal) -> float:
    """|| [[lambda; A_0..A_{m-1}]] ||_F = sqrt(lam^T (had_m A_m^T A_m) lam)
    (reference p_kruskal_norm, cpd.c:116)."""
    F = k.lam.numel()
    G = torch.ones(F, F, dtype=torch.float64, device=k.lam.device)
    for A in k.factors:
        Ad = A.double()
        G *= Ad.T @ Ad
    lam = k.lam.double()
    return float(torch.sqrt(torch.clamp(lam @ G @ lam, min=0)))


def kruskal_innerprod(k: Kruskal, t: SpTensor) -> float:
    """<X, K> = sum_nnz X_x * sum_f lam_f prod_m A_m[i_m, f]
    (reference p_tt_kruskal_inner, cpd.c:171)."""
    dev = k.lam.device
    inds = t.inds.to(dev)
    w = k.lam.double().unsqueeze(0).expand(t.nnz, -1).clone()
    for m, A in enumerate(k.factors):
        w *= A.double()[inds[m]]
    return float((w.sum(dim=1) * t.vals.to(dev).double()).sum())


def kruskal_fit(k: Kruskal, t: SpTensor) -> float:
    """fit = 1 - ||X - K||_F / ||X||_F, computed from scratch (unlike the
    in-loop fit which reuses the last MTTKRP)."""
    import math
    norm_x2 = t.vals.double().square().sum().item()
    kn = kruskal_norm(k)
    inner = kruskal_innerprod(k, t)
    residual = math.sqrt(max(0.0, norm_x2 + kn * kn - 2 * inner))
    return 1.0 - residual / math.sqrt(norm_x2)


def kruskal_to_dense(k: Kruskal) -> torch.Tensor:
    """Materialize the Kruskal tensor (small dims only)."""
    nm = len(k.factors)
    letters = "ijklmnop"[:nm]
    eq = ",".join(f"{c}f" for c in letters) + ",f->" + letters
    return torch.einsum(eq, *[A.double() for A in k.factors],
                        k.lam.double())
