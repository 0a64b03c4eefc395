"""Pure-Python simulation of the deterministic-MTTKRP walker/fixup
algorithm (csrc/hip/mttkrp_det.hip): same span math, same emission rules,
same claim/scan fixup — validates the algorithm on CPU so the GPU test
only has to confirm the HIP implementation matches it."""
import numpy as np

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
