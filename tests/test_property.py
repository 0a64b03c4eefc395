"""Property-based invariants (hypothesis) over random tensor shapes —
deepens the fixed-fixture oracle tests."""
import hypothesis.strategies as st
import torch
from hypothesis import given, settings

import splatt_amd as sp
from splatt_amd.csf import build_csf


def rand_tensor(draw):
    nm = draw(st.integers(min_value=2, max_value=5))
    dims = [draw(st.integers(min_value=2, max_value=20)) for _ in range(nm)]
    nnz = draw(st.integers(min_value=1, max_value=300))
    seed = draw(st.integers(min_value=0, max_value=2**31))
    return sp.SpTensor.synthetic(dims, nnz, seed=seed)


@settings(max_examples=25, deadline=None)
@given(data=st.data())
def test_csf_expansions_recover_sorted_coords(data):
    t = rand_tensor(data.draw)
    perm = sp.order_modes(t.dims, "smallfirst")
    c = build_csf(t, perm)
    keys = [t.inds[m] for m in perm]
    order = torch.arange(t.nnz)
    for lv in reversed(range(t.nmodes)):
        order = order[torch.argsort(keys[lv][order], stable=True)]
    for lv in range(t.nmodes):
        assert torch.equal(c.ancestor_expand(lv),
                           keys[lv][order].to(torch.int32))


@settings(max_examples=20, deadline=None)
@given(data=st.data())
def test_mttkrp_matches_oracle_any_shape(data):
    t = rand_tensor(data.draw)
    rank = data.draw(st.sampled_from([1, 3, 8]))
    mode = data.draw(st.integers(min_value=0, max_value=t.nmodes - 1))
    mats = [sp.seeded_init(d, rank, m, 5) for m, d in enumerate(t.dims)]
    cs = sp.csf_alloc(t, data.draw(st.sampled_from(["one", "two", "all"])))
    out = sp.mttkrp(cs, mats, mode)
    ref = sp.mttkrp_stream(t, mats, mode)
    assert (out - ref).abs().max() < 1e-9


@settings(max_examples=25, deadline=None)
@given(data=st.data())
def test_dedup_idempotent(data):
    t = rand_tensor(data.draw)
    f1 = t.fixed(dedup=True)
    f2 = f1.fixed(dedup=True)
    assert f2.nnz == f1.nnz
    assert abs(float(f1.vals.sum() - t.vals.sum())) < 1e-9


@settings(max_examples=30, deadline=None)
@given(weights=st.lists(st.integers(min_value=0, max_value=1000),
                        min_size=1, max_size=200),
       parts=st.integers(min_value=1, max_value=16))
def test_ccp_is_optimal_vs_bruteforce_bound(weights, parts):
    from splatt_amd._ext import native
    bounds, bn = native().partition_weighted(weights, parts)
    loads = [sum(weights[a:b]) for a, b in zip(bounds, bounds[1:])]
    assert bounds[0] == 0 and bounds[-1] == len(weights)
    assert max(loads) == bn
    # optimality lower bounds
    assert bn >= max(weights)
    assert bn >= -(-sum(weights) // parts)


@settings(max_examples=20, deadline=None)
@given(data=st.data())
def test_rows_restricted_mttkrp_tiles_exactly(data):
    """Any partition of [0, dim) into row ranges reproduces the full
    MTTKRP exactly (the chunked comm pipeline's invariant)."""
    from splatt_amd.mttkrp import mttkrp, mttkrp_rows_ok
    t = rand_tensor(data.draw)
    t = t.fixed(dedup=True)
    cs = sp.csf_alloc(t, "all")
    rank = data.draw(st.sampled_from([4, 8, 16]))
    mats = [sp.seeded_init(d, rank, m, 3) for m, d in enumerate(t.dims)]
    mode = data.draw(st.integers(min_value=0, max_value=t.nmodes - 1))
    assert mttkrp_rows_ok(cs, mode, rank)
    full = mttkrp(cs, mats, mode)
    n = t.dims[mode]
    ncuts = data.draw(st.integers(min_value=0, max_value=4))
    cuts = sorted({0, n, *(data.draw(st.integers(min_value=0, max_value=n))
                           for _ in range(ncuts))})
    out = torch.empty_like(full)
    for a, b in zip(cuts, cuts[1:]):
        mttkrp(cs, mats, mode, out=out, rows=(a, b))
    assert (out - full).abs().max().item() < 1e-12
