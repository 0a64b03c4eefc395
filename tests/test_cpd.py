"""CPD-ALS correctness: the native C++ driver and the torch driver agree;
fit is monotone-ish and a low-rank tensor is recovered (reference asserts
CPD quality via fit behavior; see cpd.c:354-371)."""
import math

import torch

import splatt_amd as sp


def lowrank_tensor(dims, rank, seed=42):
    g = torch.Generator().manual_seed(seed)
    mats = [torch.rand(d, rank, generator=g, dtype=torch.float64) for d in dims]
    dense = torch.einsum("ir,jr,kr->ijk", *mats)
    inds = (dense.abs() > -1).nonzero().T  # fully dense as COO
    vals = dense.flatten()
    return sp.SpTensor(inds, vals, list(dims))


def test_native_and_torch_drivers_agree(small3):
    o = sp.CpdOptions(max_iters=10, tolerance=0.0)
    a = sp.cpd_als_cpu_native(small3, 8, o)
    b = sp.cpd_als(small3, 8, o)
    assert a.niters == b.niters
    assert abs(a.fit - b.fit) < 1e-8


def test_lowrank_recovery():
    t = lowrank_tensor([15, 12, 10], 3)
    k = sp.cpd_als_cpu_native(t, 3, sp.CpdOptions(max_iters=100, tolerance=1e-9))
    assert k.fit > 0.999


def test_fit_trace_reasonable(small3):
    k = sp.cpd_als(small3, 8, sp.CpdOptions(max_iters=15, tolerance=0.0))
    assert len(k.fit_trace) == 15
    assert k.fit_trace[-1] >= k.fit_trace[0] - 1e-9
    assert all(math.isfinite(f) for f in k.fit_trace)


def test_seeded_init_partition_invariant():
    full = sp.seeded_init(100, 16, 1, 999)
    lo = sp.seeded_init(40, 16, 1, 999, row0=0)
    hi = sp.seeded_init(60, 16, 1, 999, row0=40)
    assert torch.equal(full, torch.cat([lo, hi]))


def test_seeded_init_deterministic():
    a = sp.seeded_init(50, 8, 0, 7)
    b = sp.seeded_init(50, 8, 0, 7)
    c = sp.seeded_init(50, 8, 0, 8)
    assert torch.equal(a, b)
    assert not torch.equal(a, c)
    assert float(a.min()) >= 0.0 and float(a.max()) < 1.0


def test_checkpoint_resume(tmp_path, small3):
    ck = str(tmp_path / "cpd.ckpt")
    full = sp.cpd_als(small3, 8, sp.CpdOptions(max_iters=6, tolerance=0.0))
    # run 3 iterations with checkpointing, then resume for 3 more
    sp.cpd_als(small3, 8, sp.CpdOptions(max_iters=3, tolerance=0.0,
                                        checkpoint_path=ck))
    resumed = sp.cpd_als(small3, 8, sp.CpdOptions(max_iters=6, tolerance=0.0,
                                                  checkpoint_path=ck,
                                                  resume=True))
    assert abs(resumed.fit - full.fit) < 1e-10
    assert resumed.niters == 6


def test_kruskal_utilities(small3):
    from splatt_amd.kruskal import (kruskal_fit, kruskal_innerprod,
                                    kruskal_norm, kruskal_to_dense)
    k = sp.cpd_als(small3, 6, sp.CpdOptions(max_iters=8, tolerance=0.0))
    # from-scratch fit must match the in-loop fit closely
    assert abs(kruskal_fit(k, small3) - k.fit) < 1e-6
    # dense reconstruction consistency
    dense = kruskal_to_dense(k)
    assert abs(float(dense.square().sum()).__pow__(0.5) - kruskal_norm(k)) < 1e-8
    # innerprod vs dense
    dref = torch.zeros(*small3.dims, dtype=torch.float64)
    dref.index_put_(tuple(small3.inds), small3.vals.double(), accumulate=True)
    assert abs(kruskal_innerprod(k, small3) - float((dense * dref).sum())) < 1e-6


def test_checkpoint_every_interval(tmp_path, small3):
    import torch
    ck = str(tmp_path / "int.ckpt")
    sp.cpd_als(small3, 6, sp.CpdOptions(max_iters=5, tolerance=0.0,
                                        checkpoint_path=ck,
                                        checkpoint_every=2))
    state = torch.load(ck, weights_only=True)
    # last multiple-of-2 iteration is it=3 (0-indexed), i.e. 4 iterations
    assert state["iteration"] == 3
    assert len(state["factors"]) == 3


def test_cpd_regularize():
    """Ridge term: reg=0 reproduces the unregularized run exactly; a
    large reg shrinks the solution (lower fit) but still converges
    (reference p_form_gram's + reg*I, matrix.c:29-83)."""
    import splatt_amd as sp
    t = sp.SpTensor.synthetic([25, 20, 30], 2000, seed=13)
    o0 = sp.CpdOptions(max_iters=8, tolerance=0.0, seed=3)
    oz = sp.CpdOptions(max_iters=8, tolerance=0.0, seed=3, regularize=0.0)
    ob = sp.CpdOptions(max_iters=8, tolerance=0.0, seed=3, regularize=10.0)
    k0 = sp.cpd_als(t, 6, o0)
    kz = sp.cpd_als(t, 6, oz)
    kb = sp.cpd_als(t, 6, ob)
    assert k0.fit == kz.fit
    assert kb.fit < k0.fit
    assert kb.fit == kb.fit  # finite
    # host reference path honors it too
    kn = sp.cpd_als_cpu_native(t, 6, ob)
    assert kn.fit < sp.cpd_als_cpu_native(t, 6, o0).fit + 1e-12


def test_checkpoint_mismatch_rejected(tmp_path, small3):
    """Resuming from a checkpoint written by a different rank/seed run
    raises a clear error instead of silently continuing."""
    import pytest as _pytest
    ck = str(tmp_path / "ck.pt")
    sp.cpd_als(small3, 8, sp.CpdOptions(max_iters=2, tolerance=0.0,
                                        checkpoint_path=ck))
    with _pytest.raises(ValueError, match="does not match"):
        sp.cpd_als(small3, 6, sp.CpdOptions(max_iters=4, tolerance=0.0,
                                            checkpoint_path=ck, resume=True))
    with _pytest.raises(ValueError, match="seed"):
        sp.cpd_als(small3, 8, sp.CpdOptions(max_iters=4, tolerance=0.0,
                                            seed=123, checkpoint_path=ck,
                                            resume=True))
