"""I/O tests (reference tests/io_test.c): text round-trip, 0/1-index
equivalence, binary round-trip incl. width down-conversion."""
import pytest
import torch

import splatt_amd as sp


def test_tns_roundtrip(tmp_path, small3):
    p = tmp_path / "t.tns"
    small3.save(p)
    t2 = sp.load(p)
    assert t2.dims == small3.dims
    assert torch.equal(t2.inds, small3.inds)
    assert (t2.vals - small3.vals).abs().max() < 1e-15


def test_zero_vs_one_indexed(tmp_path):
    t = sp.SpTensor.synthetic([10, 12, 9], 200, seed=5)
    # ensure index 0 occurs so autodetect sees a 0-based file
    t.inds[0, 0] = 0
    one = tmp_path / "one.tns"
    zero = tmp_path / "zero.tns"
    with open(one, "w") as f:
        for i in range(t.nnz):
            f.write(" ".join(str(int(t.inds[m, i]) + 1) for m in range(3))
                    + f" {float(t.vals[i]):.17g}\n")
    with open(zero, "w") as f:
        for i in range(t.nnz):
            f.write(" ".join(str(int(t.inds[m, i])) for m in range(3))
                    + f" {float(t.vals[i]):.17g}\n")
    a, b = sp.load(one), sp.load(zero)
    assert torch.equal(a.inds, b.inds)
    assert a.dims == b.dims


def test_bin_roundtrip(tmp_path, small3):
    p = tmp_path / "t.bin"
    small3.save(p)
    t2 = sp.load(p)
    assert t2.dims == small3.dims
    assert torch.equal(t2.inds, small3.inds)
    assert torch.equal(t2.vals, small3.vals)


def test_bin_width_downconvert(tmp_path, small3):
    from splatt_amd._ext import native
    p = str(tmp_path / "t32.bin")
    native().bin_write(p, small3.inds, small3.vals, list(small3.dims), 4, 4)
    t2 = sp.load(p)
    assert torch.equal(t2.inds, small3.inds)
    assert (t2.vals - small3.vals).abs().max() < 1e-6


def test_fixed_dedup_compress():
    inds = torch.tensor([[1, 1, 5, 5], [2, 2, 3, 3], [0, 0, 7, 7]])
    vals = torch.tensor([1.0, 2.0, 3.0, 4.0], dtype=torch.float64)
    t = sp.SpTensor(inds, vals, [10, 10, 10])
    f = t.fixed(dedup=True, compress=True)
    assert f.nnz == 2
    assert sorted(f.vals.tolist()) == [3.0, 7.0]
    assert f.dims == [2, 2, 2]  # empty slices removed
    assert f.indmaps[0].tolist() == [1, 5]


def test_corrupt_binary_rejected(tmp_path):
    """Truncated/garbage binary files raise clean errors, never crash."""
    import pytest
    p = tmp_path / "bad.bin"
    p.write_bytes(b"SPLATTB1" + b"\xff" * 16)   # bad widths
    with pytest.raises(RuntimeError):
        sp.load(str(p))
    p2 = tmp_path / "bad2.bin"
    p2.write_bytes(b"NOTMAGIC" + b"\x00" * 64)
    with pytest.raises(RuntimeError):
        sp.load(str(p2))
    p3 = tmp_path / "trunc.bin"
    t = sp.SpTensor.synthetic([5, 5, 5], 50, seed=1)
    t.save(str(p3))
    data = p3.read_bytes()
    p3.write_bytes(data[: len(data) // 2])
    with pytest.raises(RuntimeError):
        sp.load(str(p3))


def test_huge_dim_rejected():
    import pytest
    from splatt_amd.csf import build_csf
    import torch
    t = sp.SpTensor(torch.zeros(2, 1, dtype=torch.int64),
                    torch.ones(1, dtype=torch.float64),
                    [2, 2**33])
    with pytest.raises((ValueError, RuntimeError)):
        build_csf(t, [0, 1])


@pytest.mark.parametrize("content,msg", [
    ("1 2 x 3.0\n", "malformed line 1"),
    ("", "no nonzeros"),
    ("1 2 3 4.0\n1 2 3\n", "malformed line 2"),          # short line
    ("1 2 3 4.0\n1 2 3 4.0 5.0\n", "malformed line 2"),  # extra token
])
def test_tns_rejects_malformed(tmp_path, content, msg):
    """Garbage tokens / inconsistent arity / empty files must raise, not
    silently truncate into wrong coordinates."""
    f = tmp_path / "bad.tns"
    f.write_text(content)
    with pytest.raises(RuntimeError, match=msg):
        sp.load(str(f))


def test_tns_accepts_comments_and_blanks(tmp_path):
    f = tmp_path / "ok.tns"
    f.write_text("# header comment\n\n1 2 3 4.0\n% other comment\n2 3 1 5.0\n")
    t = sp.load(str(f))
    assert t.nnz == 2 and t.dims == [2, 3, 3]


@pytest.mark.parametrize("iw,vw", [(4, 4), (4, 8), (8, 4), (8, 8)])
def test_reference_bin_format_read(tmp_path, small3, iw, vw):
    """Files in the REFERENCE repo's .bin layout (int32 magic=BIN_COORD,
    u64 idx/val widths, then nmodes/dims/nnz + arrays at those widths;
    reference src/io.h:71-88, io.c:161-195) load transparently."""
    import struct
    p = tmp_path / "ref.bin"
    it = {4: "I", 8: "Q"}[iw]
    vt = {4: "f", 8: "d"}[vw]
    nnz = small3.nnz
    with open(p, "wb") as f:
        f.write(struct.pack("<i", 0))             # SPLATT_BIN_COORD
        f.write(struct.pack("<QQ", iw, vw))
        f.write(struct.pack(f"<{it}", 3))         # nmodes
        f.write(struct.pack(f"<3{it}", *small3.dims))
        f.write(struct.pack(f"<{it}", nnz))
        for m in range(3):
            f.write(struct.pack(f"<{nnz}{it}",
                                *small3.inds[m].tolist()))
        f.write(struct.pack(f"<{nnz}{vt}", *small3.vals.tolist()))
    t2 = sp.load(p)
    assert t2.dims == small3.dims
    assert torch.equal(t2.inds, small3.inds)
    tol = 1e-6 if vw == 4 else 0.0
    assert (t2.vals - small3.vals).abs().max() <= tol


def test_reference_bin_csf_rejected(tmp_path):
    import struct
    p = tmp_path / "csf.bin"
    with open(p, "wb") as f:
        f.write(struct.pack("<i", 1))             # SPLATT_BIN_CSF
        f.write(struct.pack("<QQ", 8, 8))
    with pytest.raises(RuntimeError, match="BIN_CSF"):
        sp.load(p)
