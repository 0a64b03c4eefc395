"""Multi-GPU decompositions over torch.distributed (RCCL on ROCm).

Public surface:
  GridDecomp / FineDecomp   — coarse / medium / fine decompositions
  grid_cpd_als / _init / _step — distributed ALS drivers
  build_shard_csf, load_shard, write_factors, comm_stats
"""
from splatt_amd.parallel.grid import (FineDecomp, GridDecomp, best_grid,
                                      comm_stats, grid_cpd_als, grid_cpd_init,
                                      grid_cpd_step, load_shard,
                                      write_factors)
from splatt_amd.parallel.dist_cpd import (build_shard_csf, dist_cpd_als,
                                          localize_shard, partition_rows)

__all__ = [
    "FineDecomp", "GridDecomp", "best_grid", "comm_stats", "grid_cpd_als",
    "grid_cpd_init", "grid_cpd_step", "load_shard", "write_factors",
    "build_shard_csf", "dist_cpd_als", "localize_shard", "partition_rows",
]
