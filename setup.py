"""In-tree build for the MI355X-native sparse tensor engine.

Two-stage build, no hipify, no dual paths:
  1. hipcc --offload-arch=gfx950 compiles csrc/hip/*.hip to objects
     (pure CDNA4 HIP; cross-compiles fine on a GPU-less box).
  2. a torch CppExtension compiles the C++17 core + pybind boundary and
     links the HIP objects against libamdhip64.

Usage: python setup.py build_ext --inplace
"""
import os
import subprocess
import sys
from pathlib import Path

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CppExtension

ROOT = Path(__file__).resolve().parent
ROCM = Path(os.environ.get("ROCM_PATH", "/opt/rocm"))
HIPCC = str(ROCM / "bin" / "hipcc")
ARCH = os.environ.get("SPLATT_GPU_ARCH", "gfx950")

HIP_SOURCES = sorted((ROOT / "csrc" / "hip").glob("*.hip"))
CORE_SOURCES = sorted((ROOT / "csrc" / "core").glob("*.cpp"))


def build_hip_objects():
    objdir = ROOT / "build" / "hip_obj"
    objdir.mkdir(parents=True, exist_ok=True)
    objs = []
    rebuilt = False
    for src in HIP_SOURCES:
        obj = objdir / (src.stem + ".o")
        if not obj.exists() or obj.stat().st_mtime < src.stat().st_mtime:
            cmd = [
                HIPCC, f"--offload-arch={ARCH}", "-O3", "-std=c++17",
                "-fPIC", "-munsafe-fp-atomics", "-c", str(src), "-o", str(obj),
            ]
            print("[hipcc]", " ".join(cmd), flush=True)
            subprocess.check_call(cmd)
            rebuilt = True
        objs.append(str(obj))
    if rebuilt:
        # setuptools does not track extra_objects mtimes: force a relink by
        # dirtying the binding TU (else a stale _C.so keeps the old kernels)
        (ROOT / "csrc" / "pybind.cpp").touch()
        for so in (ROOT / "splatt_amd").glob("_C*.so"):
            so.unlink()
    return objs


def build_native_lib(hip_objs):
    """libsplatt.so (C API, torch-free) + the `splatt` host CLI binary.
    The HIP kernel objects are linked in so splatt_cpd_als dispatches to
    the device engine (csrc/capi/capi_gpu.cpp) when a GPU is visible."""
    bindir = ROOT / "bin"
    bindir.mkdir(exist_ok=True)
    lib = bindir / "libsplatt.so"
    exe = bindir / "splatt"
    srcs = [str(p) for p in CORE_SOURCES] + [str(ROOT / "csrc/capi/capi.cpp")]
    gpu_src = ROOT / "csrc/capi/capi_gpu.cpp"
    newest = max(Path(s).stat().st_mtime
                 for s in srcs + hip_objs
                 + [str(gpu_src), str(ROOT / "csrc/capi/splatt.h"),
                    str(ROOT / "csrc/capi/splatt_main.cpp")])
    if not lib.exists() or lib.stat().st_mtime < newest:
        # capi_gpu.cpp is host-only HIP-API code: g++ with the AMD platform
        # macro (keeps one OpenMP runtime, libgomp, across libsplatt)
        gpu_obj = ROOT / "build" / "hip_obj" / "capi_gpu.o"
        subprocess.check_call(
            ["g++", "-O3", "-std=c++17", "-fPIC", "-fopenmp",
             "-D__HIP_PLATFORM_AMD__=1", f"-I{ROCM}/include",
             f"-I{ROOT}/csrc", "-c", str(gpu_src), "-o", str(gpu_obj)])
        cxx = ["g++", "-O3", "-std=c++17", "-fPIC", "-fopenmp", "-march=native",
               f"-I{ROOT}/csrc"]
        print("[g++] libsplatt.so + splatt CLI", flush=True)
        subprocess.check_call(
            cxx + ["-shared", "-o", str(lib)] + srcs + [str(gpu_obj)]
            + hip_objs + [f"-L{ROCM}/lib", "-lamdhip64",
                          f"-Wl,-rpath,{ROCM}/lib"])
        subprocess.check_call(
            cxx + ["-o", str(exe), str(ROOT / "csrc/capi/splatt_main.cpp"),
                   f"-L{bindir}", "-lsplatt", f"-Wl,-rpath,{bindir}"])


hip_objs = build_hip_objects()
build_native_lib(hip_objs)

ext = CppExtension(
    name="splatt_amd._C",
    sources=[str(p.relative_to(ROOT)) for p in CORE_SOURCES]
    + ["csrc/pybind.cpp"],
    include_dirs=[str(ROOT / "csrc")],
    extra_compile_args=["-O3", "-std=c++17", "-fopenmp", "-march=native"],
    extra_objects=hip_objs,
    extra_link_args=["-fopenmp", f"-L{ROCM}/lib", "-lamdhip64",
                     f"-Wl,-rpath,{ROCM}/lib"],
)

setup(
    name="splatt_amd",
    version="0.1.0",
    packages=["splatt_amd", "splatt_amd.ops", "splatt_amd.parallel",
              "splatt_amd.utils"],
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=False)},
)
