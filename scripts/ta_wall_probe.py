#!/usr/bin/env python3
"""TA-wall probe: per-nnz kernel cost of the v6 LDS-staged MTTKRP as the
UNSTAGED mode's working set shrinks from L2-sized to L1-resident. If the
cost is flat, the binder is the per-access L1 tag path (address
processing), not cache capacity/bandwidth — the round-2 wall argument."""
import time

import torch

import splatt_amd as sp
from splatt_amd.mttkrp import mttkrp
from splatt_amd.parallel.dist_cpd import build_shard_csf

NNZ = 76_879_419
RANK = 16
REPS = 20


def timeit(fn):
    for _ in range(4):
        fn()
    torch.cuda.synchronize()
    tic = time.time()
    for _ in range(REPS):
        fn()
    torch.cuda.synchronize()
    return (time.time() - tic) / REPS


# mode-0 output (12092 rows); staged = mode 2 (28818); unstaged = mode 1,
# whose dim sweeps from NELL's 9184 (1.15 MB) down to 128 (16 KB, L1)
for d1 in (9184, 2048, 512, 128):
    dims = [12092, d1, 28818]
    t = sp.SpTensor.synthetic(dims, NNZ, seed=3)
    cs = build_shard_csf(t.to("cuda"), dims, "all", flat_only=True,
                         stage_rank=RANK)
    mats = [sp.seeded_init(d, RANK, m, 9).cuda() for m, d in enumerate(dims)]
    out = torch.empty(dims[0], RANK, dtype=torch.float64, device="cuda")
    ms = timeit(lambda: mttkrp(cs, mats, 0, out=out)) * 1e3
    ws_kb = d1 * RANK * 8 / 1024
    print(f"unstaged dim {d1:5d} ({ws_kb:8.0f} KB working set): "
          f"{ms:7.3f} ms  {ms / NNZ * 1e9:6.2f} ps/nnz", flush=True)
