"""C API (bin/libsplatt.so) and native `splatt` CLI binary tests
(reference tests/api_test.c: opts defaults, version; CLI behavior)."""
import ctypes
import os
import subprocess

import pytest
import torch

import splatt_amd as sp

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
LIB = os.path.join(ROOT, "bin", "libsplatt.so")
EXE = os.path.join(ROOT, "bin", "splatt")

pytestmark = pytest.mark.skipif(not os.path.exists(LIB),
                                reason="native lib not built")


@pytest.fixture(scope="module")
def lib():
    L = ctypes.CDLL(LIB)
    L.splatt_default_opts.restype = ctypes.POINTER(ctypes.c_double)
    return L


def test_version(lib):
    assert lib.splatt_version_major() == 0
    assert lib.splatt_version_minor() == 1


# reference numeric slot/return values (types_config.h:103-215):
OPT_NTHREADS, OPT_TOL, OPT_REG, OPT_NITER, OPT_VERB = 0, 1, 2, 3, 4
SUCCESS = 1


def test_default_opts(lib):
    o = lib.splatt_default_opts()
    assert o[OPT_TOL] == pytest.approx(1e-5)
    assert o[OPT_NITER] == 50
    assert o[OPT_NTHREADS] == 0
    assert o[OPT_VERB] == 1     # SPLATT_VERBOSITY_LOW
    assert o[6] == 1            # SPLATT_OPTION_CSF_ALLOC = TWOMODE
    assert o[10] == 1           # SPLATT_OPTION_DECOMP = MEDIUM
    lib.splatt_free_opts(o)


def test_csf_load_and_cpd(lib, tmp_path, small3):
    tns = str(tmp_path / "t.tns").encode()
    small3.save(tns.decode())
    o = lib.splatt_default_opts()
    nmodes = ctypes.c_uint64()
    csf = ctypes.c_void_p()
    rc = lib.splatt_csf_load(tns, ctypes.byref(nmodes), ctypes.byref(csf), o)
    assert rc == SUCCESS
    assert nmodes.value == 3
    assert lib.splatt_csf_nnz(csf) == small3.nnz

    class Kruskal(ctypes.Structure):
        _fields_ = [("rank", ctypes.c_uint64),
                    ("factors", ctypes.POINTER(ctypes.c_double) * 8),
                    ("lambda_", ctypes.POINTER(ctypes.c_double)),
                    ("nmodes", ctypes.c_uint64),
                    ("dims", ctypes.c_uint64 * 8),
                    ("fit", ctypes.c_double)]

    k = Kruskal()
    o[OPT_NITER] = 5
    rc = lib.splatt_cpd_als(csf, 8, o, ctypes.byref(k))
    assert rc == SUCCESS
    assert k.nmodes == 3
    assert 0.0 <= k.fit < 1.0
    # fit must match the Python CPU driver exactly (same seed/init)
    ref = sp.cpd_als_cpu_native(small3, 8,
                                sp.CpdOptions(max_iters=5, tolerance=1e-5))
    assert abs(k.fit - ref.fit) < 1e-10
    lib.splatt_free_kruskal(ctypes.byref(k))
    lib.splatt_free_csf(csf, o)
    lib.splatt_free_opts(o)


def test_capi_mttkrp(lib, tmp_path, small3):
    tns = str(tmp_path / "t.tns").encode()
    small3.save(tns.decode())
    o = lib.splatt_default_opts()
    nmodes = ctypes.c_uint64()
    csf = ctypes.c_void_p()
    assert lib.splatt_csf_load(tns, ctypes.byref(nmodes),
                               ctypes.byref(csf), o) == SUCCESS
    rank = 8
    mats = [sp.seeded_init(d, rank, m, 1).contiguous()
            for m, d in enumerate(small3.dims)]
    ptrs = (ctypes.POINTER(ctypes.c_double) * 3)(
        *[ctypes.cast(m.data_ptr(), ctypes.POINTER(ctypes.c_double))
          for m in mats])
    out = torch.zeros(small3.dims[1], rank, dtype=torch.float64)
    rc = lib.splatt_mttkrp(1, rank, csf, ptrs,
                           ctypes.cast(out.data_ptr(),
                                       ctypes.POINTER(ctypes.c_double)), o)
    assert rc == SUCCESS
    ref = sp.mttkrp_stream(small3, mats, 1)
    assert (out - ref).abs().max() < 1e-10
    lib.splatt_free_csf(csf, o)
    lib.splatt_free_opts(o)


def test_cli_binary(tmp_path, small3):
    tns = str(tmp_path / "t.tns")
    small3.save(tns)
    r = subprocess.run([EXE, "stats", tns], capture_output=True, text=True)
    assert r.returncode == 0 and "NNZ" in r.stdout
    r = subprocess.run([EXE, "cpd", tns, "-r", "4", "-i", "3", "--nowrite"],
                       capture_output=True, text=True, cwd=tmp_path)
    assert r.returncode == 0 and "Final fit" in r.stdout
    r = subprocess.run([EXE, "convert", tns, str(tmp_path / "t.bin")],
                       capture_output=True, text=True)
    assert r.returncode == 0
    r = subprocess.run([EXE, "stats", str(tmp_path / "t.bin")],
                       capture_output=True, text=True)
    assert r.returncode == 0 and "NNZ" in r.stdout


def test_csf_convert_in_memory(lib, small3):
    """splatt_csf_convert from raw index/value arrays."""
    import numpy as np
    nnz = small3.nnz
    inds = [small3.inds[m].numpy().astype(np.uint64).copy() for m in range(3)]
    vals = small3.vals.double().numpy().copy()
    arr_t = ctypes.POINTER(ctypes.c_uint64) * 3
    ptrs = arr_t(*[i.ctypes.data_as(ctypes.POINTER(ctypes.c_uint64))
                   for i in inds])
    o = lib.splatt_default_opts()
    csf = ctypes.c_void_p()
    rc = lib.splatt_csf_convert(3, nnz, ptrs,
                                vals.ctypes.data_as(
                                    ctypes.POINTER(ctypes.c_double)),
                                ctypes.byref(csf), o)
    assert rc == SUCCESS
    assert lib.splatt_csf_nnz(csf) == nnz
    lib.splatt_free_csf(csf, o)
    lib.splatt_free_opts(o)


def test_cli_binary_missing_file():
    r = subprocess.run([EXE, "stats", "/nonexistent.tns"],
                       capture_output=True, text=True)
    assert r.returncode == 1
    assert "cannot open" in r.stderr


def test_capi_example_compiles_and_runs(tmp_path):
    """examples/capi_demo.c must build against csrc/capi/splatt.h and run
    end to end (load -> CPD -> MTTKRP) on a small tensor."""
    t = sp.SpTensor.synthetic([20, 15, 25], 800, seed=6).fixed()
    tns = tmp_path / "t.tns"
    t.save(tns)
    exe = tmp_path / "capi_demo"
    r = subprocess.run(
        ["gcc", "-O2", os.path.join(ROOT, "examples", "capi_demo.c"),
         "-I" + os.path.join(ROOT, "csrc", "capi"),
         "-L" + os.path.join(ROOT, "bin"), "-lsplatt",
         "-Wl,-rpath," + os.path.join(ROOT, "bin"), "-o", str(exe)],
        capture_output=True, text=True)
    assert r.returncode == 0, r.stderr[-500:]
    r = subprocess.run([str(exe), str(tns), "6"], capture_output=True,
                       text=True, timeout=120)
    assert r.returncode == 0, (r.stdout, r.stderr[-300:])
    assert "CPD fit:" in r.stdout and "mttkrp mode 0" in r.stdout


def test_cli_binary_reorder(tmp_path):
    """Native `splatt reorder`: random relabeling preserves shape/values;
    --type perm re-applies a written permutation reproducibly."""
    t = sp.SpTensor.synthetic([12, 9, 15], 300, seed=4).fixed()
    tns = tmp_path / "t.tns"
    t.save(tns)
    out1 = tmp_path / "r1.tns"
    out2 = tmp_path / "r2.tns"
    pfx = str(tmp_path / "p")
    r = subprocess.run([EXE, "reorder", str(tns), str(out1), "--seed", "9",
                        "--permfile", pfx], capture_output=True, text=True)
    assert r.returncode == 0, r.stderr
    r = subprocess.run([EXE, "reorder", str(tns), str(out2), "--type",
                        "perm", "--permfile", pfx],
                       capture_output=True, text=True)
    assert r.returncode == 0, r.stderr
    assert out1.read_text() == out2.read_text()
    t1 = sp.load(str(out1))
    assert t1.dims == t.dims and t1.nnz == t.nnz
    assert torch.allclose(t1.vals.sort().values, t.vals.sort().values)
    for m in range(3):
        assert torch.equal(torch.bincount(t1.inds[m]).sort().values,
                           torch.bincount(t.inds[m]).sort().values)


def test_capi_regularize_option(lib, tmp_path):
    """SPLATT_OPTION_REGULARIZE flows into the C API CPD."""
    t = sp.SpTensor.synthetic([20, 18, 22], 900, seed=15).fixed()
    tns = tmp_path / "t.tns"
    t.save(tns)
    out = subprocess.run([EXE, "cpd", str(tns), "-r", "5", "-i", "6",
                          "--nowrite"], capture_output=True, text=True)
    base = float(out.stdout.split("Final fit:")[1].split()[0])
    out = subprocess.run([EXE, "cpd", str(tns), "-r", "5", "-i", "6",
                          "--reg", "25.0", "--nowrite"],
                         capture_output=True, text=True)
    reg = float(out.stdout.split("Final fit:")[1].split()[0])
    assert reg < base


@pytest.mark.gpu
def test_capi_cpd_uses_gpu_engine(lib, tmp_path):
    """On a GPU box, splatt_cpd_als dispatches to the HIP engine
    (csrc/capi/capi_gpu.cpp) and its fit matches the Python device
    driver (reference include/splatt.h C surface)."""
    import torch
    assert torch.cuda.is_available()
    assert lib.splatt_gpu_available() == 1
    t = sp.SpTensor.synthetic([300, 250, 400], 60_000, seed=11).fixed()
    tns = str(tmp_path / "t.tns").encode()
    t.save(tns.decode())
    o = lib.splatt_default_opts()
    nmodes = ctypes.c_uint64()
    csf = ctypes.c_void_p()
    assert lib.splatt_csf_load(tns, ctypes.byref(nmodes),
                               ctypes.byref(csf), o) == SUCCESS

    class Kruskal(ctypes.Structure):
        _fields_ = [("rank", ctypes.c_uint64),
                    ("factors", ctypes.POINTER(ctypes.c_double) * 8),
                    ("lambda_", ctypes.POINTER(ctypes.c_double)),
                    ("nmodes", ctypes.c_uint64),
                    ("dims", ctypes.c_uint64 * 8),
                    ("fit", ctypes.c_double)]

    k = Kruskal()
    o[OPT_NITER] = 5
    o[OPT_TOL] = 0.0
    assert lib.splatt_cpd_als(csf, 16, o, ctypes.byref(k)) == SUCCESS
    # Python device driver, same seed/policy/schedule
    cs = sp.csf_alloc(t.to("cuda"), "two")
    ref = sp.cpd_als(cs, 16, sp.CpdOptions(max_iters=5, tolerance=0.0))
    assert abs(k.fit - ref.fit) < 1e-6, (k.fit, ref.fit)
    # and the CPU C path agrees too (proves the dispatch changed engines,
    # not the math)
    k2 = Kruskal()
    import os as _os
    _os.environ["SPLATT_CAPI_CPU"] = "1"
    try:
        assert lib.splatt_cpd_als(csf, 16, o, ctypes.byref(k2)) == SUCCESS
    finally:
        del _os.environ["SPLATT_CAPI_CPU"]
    assert abs(k.fit - k2.fit) < 1e-6, (k.fit, k2.fit)
    lib.splatt_free_kruskal(ctypes.byref(k))
    lib.splatt_free_kruskal(ctypes.byref(k2))
    lib.splatt_free_csf(csf, o)
    lib.splatt_free_opts(o)


def test_mex_sources_compile(tmp_path):
    """The Octave/MATLAB MEX bindings (matlab/*.c) must be valid C
    against the public header — compile-checked with a stub mex.h since
    no MEX toolchain ships in this image (matlab/README.md)."""
    import glob
    srcs = sorted(glob.glob(os.path.join(ROOT, "matlab", "*.c")))
    assert len(srcs) >= 4
    for src in srcs:
        r = subprocess.run(
            ["gcc", "-fsyntax-only", "-Wall", "-Werror",
             "-I" + os.path.join(ROOT, "tests", "mex_stub"),
             "-I" + os.path.join(ROOT, "csrc", "capi"), src],
            capture_output=True, text=True)
        assert r.returncode == 0, (src, r.stderr[-800:])


@pytest.mark.gpu
def test_cli_binary_gpu_engine(tmp_path):
    """The native `splatt` binary served by the HIP engine end to end
    (splatt_cpd_als GPU dispatch behind the C CLI)."""
    t = sp.SpTensor.synthetic([300, 250, 400], 60_000, seed=11).fixed()
    tns = str(tmp_path / "t.tns")
    t.save(tns)
    r = subprocess.run([EXE, "cpd", tns, "-r", "16", "-i", "5", "-t", "0",
                        "--nowrite"], capture_output=True, text=True,
                       cwd=tmp_path)
    assert r.returncode == 0, r.stderr[-500:]
    fit = float(r.stdout.split("Final fit:")[1].split()[0])
    cs = sp.csf_alloc(t.to("cuda"), "two")
    ref = sp.cpd_als(cs, 16, sp.CpdOptions(max_iters=5, tolerance=0.0))
    assert abs(fit - ref.fit) < 1e-4, (fit, ref.fit)


@pytest.mark.gpu
def test_capi_mttkrp_gpu_engine(lib, tmp_path):
    """splatt_mttkrp through the HIP engine with the device-CSF cache on
    the handle (second call reuses the uploaded streams)."""
    t = sp.SpTensor.synthetic([300, 250, 400], 60_000, seed=13).fixed()
    tns = str(tmp_path / "t.tns").encode()
    t.save(tns.decode())
    o = lib.splatt_default_opts()
    nmodes = ctypes.c_uint64()
    csf = ctypes.c_void_p()
    assert lib.splatt_csf_load(tns, ctypes.byref(nmodes),
                               ctypes.byref(csf), o) == SUCCESS
    rank = 16
    mats = [sp.seeded_init(d, rank, m, 2).contiguous()
            for m, d in enumerate(t.dims)]
    ptrs = (ctypes.POINTER(ctypes.c_double) * 3)(
        *[ctypes.cast(m.data_ptr(), ctypes.POINTER(ctypes.c_double))
          for m in mats])
    for mode in range(3):        # repeated calls exercise the cache
        out = torch.zeros(t.dims[mode], rank, dtype=torch.float64)
        rc = lib.splatt_mttkrp(mode, rank, csf, ptrs,
                               ctypes.cast(out.data_ptr(),
                                           ctypes.POINTER(ctypes.c_double)),
                               o)
        assert rc == SUCCESS
        ref = sp.mttkrp_stream(t, mats, mode)
        assert (out - ref).abs().max() < 1e-8, mode
    lib.splatt_free_csf(csf, o)
    lib.splatt_free_opts(o)
