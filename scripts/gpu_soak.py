#!/usr/bin/env python3
"""GPU soak: random tensor/config trials, every MTTKRP checked against
the CPU oracle; deterministic-mode trials additionally checked for
torch.equal across two runs. Usage: gpu_soak.py [trials] [seed]."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import splatt_amd as sp


def main() -> int:
    trials = int(sys.argv[1]) if len(sys.argv) > 1 else 60
    seed = int(sys.argv[2]) if len(sys.argv) > 2 else 0
    g = torch.Generator().manual_seed(seed)

    def ri(lo, hi):
        return int(torch.randint(lo, hi + 1, (1,), generator=g))

    t0 = time.time()
    fails = 0
    for i in range(trials):
        nm = ri(3, 5)
        dims = [ri(8, 3000) for _ in range(nm)]
        nnz = ri(2_000, 200_000)
        rank = [4, 8, 16, 32, 64, 7, 20][ri(0, 6)]
        policy = ["one", "two", "all"][ri(0, 2)]
        det = policy == "all" and rank in (4, 8, 16, 32, 64) and ri(0, 1)
        dist = ["uniform", "zipf"][ri(0, 1)]
        t = sp.SpTensor.synthetic(dims, nnz, seed=seed * 1000 + i, dist=dist)
        mats_c = [sp.seeded_init(d, rank, m, 99 + i) for m, d in
                  enumerate(t.dims)]
        mats_g = [m.cuda() for m in mats_c]
        cs = sp.csf_alloc(t.to("cuda"), policy)
        for mode in range(nm):
            out = sp.mttkrp(cs, mats_g, mode,
                            deterministic=bool(det) or None)
            ref = sp.mttkrp_stream(t, mats_c, mode)
            err = (out.cpu() - ref).abs().max().item()
            scale = ref.abs().max().item() or 1.0
            if err / scale > 1e-10:
                fails += 1
                print(f"FAIL trial {i} mode {mode}: cfg=({dims},{nnz},"
                      f"r{rank},{policy},det={det},{dist}) err={err:.3e}")
            if det:
                again = sp.mttkrp(cs, mats_g, mode, deterministic=True)
                if not torch.equal(out, again):
                    fails += 1
                    print(f"FAIL det-repeat trial {i} mode {mode}")
        del cs, mats_g
        torch.cuda.empty_cache()
    print(f"gpu soak: {trials} trials, {fails} failures, "
          f"{time.time() - t0:.0f}s")
    return 1 if fails else 0


if __name__ == "__main__":
    sys.exit(main())
