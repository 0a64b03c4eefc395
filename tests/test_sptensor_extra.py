"""Synthetic generator distributions + big-rank GPU fallback."""
import torch

import splatt_amd as sp


def test_zipf_synthetic_is_skewed():
    t = sp.SpTensor.synthetic([500, 400, 600], 50_000, seed=3, dist="zipf")
    assert t.nnz == 50_000
    assert all(int(t.inds[m].max()) < t.dims[m] for m in range(3))
    # heavy-tailed: top slice much larger than mean occupancy
    h = torch.bincount(t.inds[0], minlength=500)
    assert float(h.max()) > 10 * float(h.float().mean())
    # and MTTKRP still matches the oracle on it
    mats = [sp.seeded_init(d, 8, m, 5) for m, d in enumerate(t.dims)]
    cs = sp.csf_alloc(t, "two")
    out = sp.mttkrp(cs, mats, 1)
    ref = sp.mttkrp_stream(t, mats, 1)
    assert (out - ref).abs().max() < 1e-9


def test_unfold_matches_dense():
    import torch
    import splatt_amd as sp
    t = sp.SpTensor.synthetic([7, 5, 6], 80, seed=3)
    dense = torch.zeros(*t.dims, dtype=t.vals.dtype)
    dense.index_put_((t.inds[0], t.inds[1], t.inds[2]), t.vals,
                     accumulate=True)   # synthetic may contain duplicates
    for mode in range(3):
        X = t.unfold(mode).to_dense()
        others = [m for m in range(3) if m != mode]
        ref = dense.permute(mode, *others).reshape(t.dims[mode], -1)
        assert torch.allclose(X, ref), mode
