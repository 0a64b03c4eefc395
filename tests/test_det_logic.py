"""Pure-Python simulation of the deterministic-MTTKRP walker/fixup
algorithm (csrc/hip/mttkrp_det.hip): same span math, same emission rules,
same claim/scan fixup — validates the algorithm on CPU so the GPU test
only has to confirm the HIP implementation matches it."""
import random

import numpy as np
import pytest

WAVE = 64


def pick_span(nnz, target=65536):
    return min(16384, max(256, nnz // target))


def walker_range(w, nnz, span, R):
    wid, g = divmod(w, R)
    w0 = wid * span
    if w0 >= nnz:
        return None
    w1 = min(nnz, w0 + span)
    gsz = (w1 - w0 + R - 1) // R
    p0 = min(w1, w0 + g * gsz)
    p1 = min(w1, p0 + gsz)
    return (p0, p1) if p0 < p1 else None


def det_sim(key, x, rank):
    """Simulate kernel + fixup; x: per-position scalar contribution."""
    nnz = len(key)
    span = pick_span(nnz)
    R = WAVE // rank
    NW = ((nnz + span - 1) // span) * R
    nkeys = int(key.max()) + 1
    out = np.zeros(nkeys)
    side = np.zeros((NW, 2))
    claimed = []
    for w in range(NW):                      # main kernel
        r = walker_range(w, nnz, span, R)
        if r is None:
            continue
        p0, p1 = r
        kf, kl = key[p0], key[p1 - 1]
        cur, acc = kf, 0.0
        for p in range(p0, p1):
            if key[p] != cur:
                if cur == kf:
                    side[w, 0] = acc
                else:
                    out[cur] = acc           # interior: plain store
                acc = 0.0
                cur = key[p]
            acc += x[p]
        if cur == kf:
            side[w, 0] = acc
        elif cur == kl:
            side[w, 1] = acc
        else:
            out[cur] = acc
    for w in range(NW):                      # fixup
        r = walker_range(w, nnz, span, R)
        if r is None:
            continue
        p0, p1 = r
        kf, kl = key[p0], key[p1 - 1]

        def resolve(k, w0=w):
            tot = 0.0
            for xw in range(w0, NW):
                rr = walker_range(xw, nnz, span, R)
                if rr is None:
                    continue
                q0, q1 = rr
                xf, xl = key[q0], key[q1 - 1]
                if xf > k:
                    break
                if xf == k:
                    tot += side[xw, 0]
                if xl == k:
                    tot += side[xw, 1]
                if xl > k:
                    break
            out[k] = tot
            claimed.append(k)

        if p0 == 0 or key[p0 - 1] != kf:
            resolve(kf)
        if kl != kf:
            resolve(kl)
    assert len(claimed) == len(set(claimed)), "boundary key claimed twice"
    return out


def _check(key, x, rank, label):
    key = np.asarray(key, dtype=np.int64)
    x = np.asarray(x, dtype=np.float64)
    ref = np.zeros(int(key.max()) + 1)
    np.add.at(ref, key, x)
    got = det_sim(key, x, rank)
    assert np.allclose(got, ref, rtol=1e-12, atol=1e-12), \
        (label, np.abs(got - ref).max())


def test_det_walker_fixup_logic():
    rng = np.random.default_rng(0)
    for rank in (16, 64):
        n = 300_000
        key = np.sort(rng.integers(0, 2000, n))
        _check(key, rng.standard_normal(n), rank, f"r{rank} powerlaw")
        # one giant run spanning many walkers
        key = np.sort(np.concatenate([np.zeros(150_000, np.int64),
                                      rng.integers(1, 50, 150_000)]))
        _check(key, rng.standard_normal(300_000), rank, f"r{rank} giant-run")
        # every key distinct (all interior or trivial boundaries)
        n = 70_000
        _check(np.arange(n), rng.standard_normal(n), rank, f"r{rank} uniq")
        # single walker
        _check(np.sort(rng.integers(0, 5, 100)),
               rng.standard_normal(100), rank, f"r{rank} tiny")
        # runs aligned to span boundaries
        key = np.repeat(np.arange(1200), 64)[:300_000]
        _check(key, rng.standard_normal(len(key)), rank, f"r{rank} aligned")


def test_det_walker_fixup_random():
    """Randomized shapes: random run-length distributions, ranks and
    sizes — claim uniqueness + exact totals must hold for all."""
    rng = np.random.default_rng(7)
    for trial in range(30):
        rank = int(rng.choice([4, 8, 16, 32, 64]))
        n = int(rng.integers(1, 40_000))
        style = trial % 3
        if style == 0:
            key = np.sort(rng.integers(0, max(1, n // 50) + 1, n))
        elif style == 1:   # few giant runs
            key = np.sort(rng.integers(0, 4, n))
        else:              # zipf-ish run lengths
            key = np.sort(rng.zipf(1.3, n) % max(2, n // 100))
        _check(key, rng.standard_normal(n), rank, f"trial{trial} r{rank}")


def _det6_simulate(key, bucket_of, blk, chunkrows, F_lanes, vals):
    """Pure-Python mirror of the det6 kernel's store discipline
    (csrc/hip/mttkrp_det.hip mttkrp_det6_kern/_fixup/_fold): walker spans
    derived from block descriptors (NSUB sub-spans per block), plain
    stores for walker-interior runs into per-bucket outputs, side-buffer
    boundaries resolved by the ordered fixup, ascending-bucket fold.
    Values here are per-nnz contributions; output is per-key sums."""
    NSUB = 4 * (64 // F_lanes)
    starts, ends, row0s, bkt0s = blk
    nblocks = len(starts)
    nbuckets = max(bucket_of) + 1 if bucket_of else 1
    outb = [{} for _ in range(nbuckets)]
    side = {}

    def walker_range(w):
        b = w // NSUB
        s = w % NSUB
        a0, a1 = starts[b], ends[b]
        gsz = -(-(a1 - a0) // NSUB)
        p0 = min(a1, a0 + s * gsz)
        p1 = min(a1, p0 + gsz)
        return (p0, p1, b) if p0 < p1 else None

    nwalk = nblocks * NSUB
    for w in range(nwalk):
        r = walker_range(w)
        if r is None:
            continue
        p0, p1, b = r
        bucket = row0s[b] // chunkrows
        kf, kl = key[p0], key[p1 - 1]
        cur, acc = kf, 0.0
        for p in range(p0, p1):
            if key[p] != cur:
                if cur == kf:
                    side[(w, 0)] = acc
                else:
                    outb[bucket][cur] = acc
                acc = 0.0
                cur = key[p]
            acc += vals[p]
        if cur == kf:
            side[(w, 0)] = acc
        elif cur == kl:
            side[(w, 1)] = acc
        else:
            outb[bucket][cur] = acc

    for w in range(nwalk):
        r = walker_range(w)
        if r is None:
            continue
        p0, p1, b = r
        myrow0 = row0s[b]
        bucket = myrow0 // chunkrows
        kf, kl = key[p0], key[p1 - 1]

        def resolve(k):
            tot = 0.0
            for x in range(w, nwalk):
                rx = walker_range(x)
                if rx is None:
                    continue
                q0, q1, bx = rx
                if row0s[bx] != myrow0:
                    break
                xf, xl = key[q0], key[q1 - 1]
                if xf > k:
                    break
                if xf == k:
                    tot += side.get((x, 0), 0.0)
                if xl == k:
                    tot += side.get((x, 1), 0.0)
                if xl > k:
                    break
            outb[bucket][k] = tot

        is_first = (p0 == bkt0s[b]) or (key[p0 - 1] != kf)
        if is_first:
            resolve(kf)
        if kl != kf:
            resolve(kl)

    out = {}
    for b in range(nbuckets):
        for k, v in outb[b].items():
            out[k] = out.get(k, 0.0) + v
    return out


@pytest.mark.parametrize("seed", [0, 1, 2, 3])
@pytest.mark.parametrize("F_lanes", [16, 64])
def test_det6_store_discipline_simulation(seed, F_lanes):
    """Randomized bucket-major streams through the det6 simulation must
    reproduce plain per-key accumulation exactly."""
    rng = random.Random(seed)
    nbuckets = rng.randint(1, 6)
    chunkrows = rng.randint(2, 9)
    key, bucket_of, vals = [], [], []
    for b in range(nbuckets):
        nseg = rng.randint(0, 400)
        ks = sorted(rng.randint(0, 60) for _ in range(nseg))
        key += ks
        bucket_of += [b] * nseg
        vals += [rng.uniform(-1, 1) for _ in range(nseg)]
    # block descriptors: bucket segments split into tgt-size blocks
    starts, ends, row0s, bkt0s = [], [], [], []
    tgt = rng.randint(3, 50)
    p = 0
    for b in range(nbuckets):
        e = p + bucket_of.count(b)
        s = p
        q = s
        while q < e:
            r = min(e, q + tgt)
            starts.append(q)
            ends.append(r)
            row0s.append(b * chunkrows)
            bkt0s.append(s)
            q = r
        p = e
    got = _det6_simulate(key, bucket_of, (starts, ends, row0s, bkt0s),
                         chunkrows, F_lanes, vals)
    want = {}
    for k, v in zip(key, vals):
        want[k] = want.get(k, 0.0) + v
    assert set(got) == set(want)
    for k in want:
        assert abs(got[k] - want[k]) < 1e-9, (k, got[k], want[k])
