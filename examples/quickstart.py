"""End-to-end walkthrough of the splatt_amd API.

Runs on CPU; put tensors on "cuda" for the gfx950 HIP path.
    python examples/quickstart.py
"""
import os as _os
import sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))
import torch

import splatt_amd as sp


def main():
    # 1. a sparse tensor: load a .tns/.bin file, or synthesize one
    t = sp.SpTensor.synthetic([200, 150, 300], 50_000, seed=7, dist="zipf")
    print("tensor:", t.dims, t.nnz, "nnz")

    # 2. repair pipeline (the `splatt check` operations)
    t = t.fixed(dedup=True)

    # 3. CSF allocation (one | two | all copies) + device residency
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    cs = sp.csf_alloc(t.to(dev), "all")
    print("csf copies:", len(cs.csfs), "bytes:", cs.storage_bytes())

    # 4. one MTTKRP against the gold oracle
    mats = [sp.seeded_init(d, 16, m, 42).to(dev) for m, d in enumerate(t.dims)]
    out = sp.mttkrp(cs, mats, mode=0)
    ref = sp.mttkrp_stream(t, [m.cpu() for m in mats], 0)
    print("mttkrp max err vs oracle:", float((out.cpu() - ref).abs().max()))

    # 5. CPD-ALS (device-resident when dev == cuda)
    k = sp.cpd_als(cs, rank=16, opts=sp.CpdOptions(max_iters=20))
    print(f"fit = {k.fit:.5f} after {k.niters} iterations")

    # 6. Kruskal utilities
    print("||K|| =", sp.kruskal_norm(k))
    print("fit (from scratch) =", sp.kruskal_fit(k, t))


if __name__ == "__main__":
    main()
