"""Oracle cross-validation of MTTKRP — the reference's central test pattern
(tests/mttkrp_test.c:36-250): mttkrp_stream (COO) is gold; every CSF
configuration must match it elementwise, plus a dense einsum reference."""
import pytest
import torch

import splatt_amd as sp

POLICIES = ["one", "two", "all"]


def dense_mttkrp(t: sp.SpTensor, mats, mode):
    dense = torch.zeros(*t.dims, dtype=torch.float64)
    dense.index_put_(tuple(t.inds), t.vals.double(), accumulate=True)
    letters = "ijklm"[: t.nmodes]
    rhs = ",".join(f"{letters[m]}f" for m in range(t.nmodes) if m != mode)
    eq = f"{letters},{rhs}->{letters[mode]}f"
    others = [mats[m].double() for m in range(t.nmodes) if m != mode]
    return torch.einsum(eq, dense, *others)


def make_mats(dims, rank, seed=123):
    return [sp.seeded_init(d, rank, m, seed) for m, d in enumerate(dims)]


def test_stream_matches_dense(small3):
    mats = make_mats(small3.dims, 16)
    for mode in range(3):
        out = sp.mttkrp_stream(small3, mats, mode)
        ref = dense_mttkrp(small3, mats, mode)
        assert (out - ref).abs().max() < 1e-10


@pytest.mark.parametrize("policy", POLICIES)
@pytest.mark.parametrize("rank", [3, 16])
def test_csf_matches_stream_3mode(small3, policy, rank):
    mats = make_mats(small3.dims, rank)
    cs = sp.csf_alloc(small3, policy)
    for mode in range(3):
        out = sp.mttkrp(cs, mats, mode)
        ref = sp.mttkrp_stream(small3, mats, mode)
        assert (out - ref).abs().max() < 1e-10, (policy, mode, rank)


@pytest.mark.parametrize("policy", POLICIES)
def test_csf_matches_stream_4mode(med4, policy):
    mats = make_mats(med4.dims, 8)
    cs = sp.csf_alloc(med4, policy)
    for mode in range(4):
        out = sp.mttkrp(cs, mats, mode)
        ref = sp.mttkrp_stream(med4, mats, mode)
        assert (out - ref).abs().max() < 1e-10, (policy, mode)


def test_csf_matches_stream_5mode(med5):
    mats = make_mats(med5.dims, 5)
    cs = sp.csf_alloc(med5, "two")
    for mode in range(5):
        out = sp.mttkrp(cs, mats, mode)
        ref = sp.mttkrp_stream(med5, mats, mode)
        assert (out - ref).abs().max() < 1e-10


def test_f32_path(small3):
    t32 = sp.SpTensor(small3.inds, small3.vals.float(), small3.dims)
    mats = [m.float() for m in make_mats(t32.dims, 16)]
    cs = sp.csf_alloc(t32, "two")
    for mode in range(3):
        out = sp.mttkrp(cs, mats, mode)
        ref = dense_mttkrp(small3, [m.double() for m in mats], mode)
        assert (out.double() - ref).abs().max() < 5e-3


def test_privatized_cpu_path():
    """Short output modes trigger the privatized (replicated-output)
    walker; must match the oracle exactly (reference p_is_privatized)."""
    t = sp.SpTensor.synthetic([200, 6, 150], 60_000, seed=41)
    mats = make_mats(t.dims, 8)
    cs = sp.csf_alloc(t, "one")  # mode 1 lands mid/leaf, tiny dim
    for mode in range(3):
        out = sp.mttkrp(cs, mats, mode, nthreads=7)
        ref = sp.mttkrp_stream(t, mats, mode)
        assert (out - ref).abs().max() < 1e-10, mode


def test_degenerate_gram_survives():
    """Rank-deficient normal equations (duplicate factor columns) must not
    crash — Tikhonov escalation (reference gelss fallback analog)."""
    t = sp.SpTensor.synthetic([30, 25, 35], 3000, seed=9)
    k = sp.cpd_als_cpu_native(t, 24, sp.CpdOptions(max_iters=4, tolerance=0.0))
    assert all(abs(f) < 1.0 + 1e-9 for f in [k.fit])
