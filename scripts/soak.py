"""Randomized oracle soak across builders / ranks / mode counts.

CPU by default; on a GPU box every device builder (csf_alloc, flat-only,
LDS-staged) is exercised. Used ad hoc; the bounded version lives in
tests/test_property.py.
    python scripts/soak.py [ntrials]
"""
import random
import sys

import torch

sys.path.insert(0, __file__.rsplit("/", 2)[0])
import splatt_amd as sp  # noqa: E402
from splatt_amd.parallel.dist_cpd import build_shard_csf  # noqa: E402


def main(ntrials=100):
    gpu = torch.cuda.is_available()
    rng = random.Random(torch.initial_seed() & 0x7FFFFFFF)
    fails = 0
    for trial in range(ntrials):
        nm = rng.randint(2, 5)
        dims = [rng.randint(5, 600 if gpu else 60) for _ in range(nm)]
        nnz = rng.randint(100, 150_000 if gpu else 2_000)
        t = sp.SpTensor.synthetic(dims, nnz, seed=rng.randint(0, 2**31),
                                  dist=rng.choice(["uniform", "zipf"]))
        rank = rng.choice([1, 4, 8, 16, 32, 64, 7, 100])
        mode = rng.randrange(nm)
        if gpu and nm >= 3:
            td = t.to("cuda")
            kind = rng.choice(["alloc", "flat", "staged"])
            if kind == "alloc":
                cs = sp.csf_alloc(td, rng.choice(["one", "two", "all"]))
            elif kind == "flat":
                cs = build_shard_csf(td, list(dims), "all", flat_only=True)
            else:
                cs = build_shard_csf(td, list(dims), "all", flat_only=True,
                                     stage_rank=rank if rank <= 64 else 0)
        else:
            cs = sp.csf_alloc(t, rng.choice(["one", "two", "all"]))
        mats = [sp.seeded_init(d, rank, m, trial) for m, d in enumerate(dims)]
        dev_mats = [m.cuda() for m in mats] if gpu and nm >= 3 else mats
        out = sp.mttkrp(cs, dev_mats, mode)
        ref = sp.mttkrp_stream(t, mats, mode)
        err = float((out.cpu() - ref).abs().max())
        if err > 1e-7:
            fails += 1
            print("FAIL", dims, nnz, rank, mode, err)
    print(f"soak: {ntrials} trials, {fails} failures")
    return 1 if fails else 0


if __name__ == "__main__":
    sys.exit(main(int(sys.argv[1]) if len(sys.argv) > 1 else 100))
