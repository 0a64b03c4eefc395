"""splatt_amd — an MI355X-native sparse tensor factorization engine.

A from-scratch rebuild of the capabilities of SPLATT (the Surprisingly
ParalleL spArse Tensor Toolkit): CPD via ALS around a fast CSF MTTKRP —
designed CDNA4-first: hand-written gfx950 HIP kernels for the sparse hot
path, rocBLAS (through PyTorch-ROCm) for the dense normal equations, RCCL
over xGMI (torch.distributed) for multi-GPU decompositions.
"""

__version__ = "0.1.0"
VERSION = (0, 1, 0)

from splatt_amd.sptensor import SpTensor
from splatt_amd.csf import Csf, CsfSet, CsfAllocPolicy, build_csf, csf_alloc, order_modes
from splatt_amd.mttkrp import mttkrp, mttkrp_stream
from splatt_amd.cpd import CpdOptions, Kruskal, cpd_als, cpd_als_cpu_native, seeded_init
from splatt_amd.kruskal import (kruskal_fit, kruskal_innerprod, kruskal_norm,
                                kruskal_to_dense)

load = SpTensor.load

__all__ = [
    "SpTensor", "Csf", "CsfSet", "CsfAllocPolicy", "build_csf", "csf_alloc",
    "order_modes", "mttkrp", "mttkrp_stream", "CpdOptions", "Kruskal",
    "cpd_als", "cpd_als_cpu_native", "seeded_init", "load",
    "kruskal_fit", "kruskal_innerprod", "kruskal_norm", "kruskal_to_dense",
]
