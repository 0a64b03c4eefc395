#!/usr/bin/env bash
# Sanitizer pass over the torch-free C++ core + native CLI:
# ASan + UBSan + LeakSanitizer across every subcommand.
set -euo pipefail
cd "$(dirname "$0")/.."
g++ -std=c++17 -O1 -g -fopenmp -fsanitize=address,undefined \
    -fno-omit-frame-pointer -Icsrc \
    -D__HIP_PLATFORM_AMD__=1 -I"${ROCM_PATH:-/opt/rocm}/include" \
    csrc/core/*.cpp csrc/capi/capi.cpp csrc/capi/capi_gpu.cpp \
    csrc/capi/splatt_main.cpp \
    build/hip_obj/dense_kernels.o build/hip_obj/mttkrp_det.o \
    build/hip_obj/mttkrp_flat.o build/hip_obj/mttkrp_kernels.o \
    build/hip_obj/mttkrp_lds.o \
    -L"${ROCM_PATH:-/opt/rocm}/lib" -lamdhip64 \
    -Wl,-rpath,"${ROCM_PATH:-/opt/rocm}/lib" \
    -o /tmp/splatt_asan
# (the HIP kernel objects are linked un-sanitized; on a GPU-less box
# capi_gpu_available() is false and the sanitized CPU core serves every
# call)
python - <<'PY'
import splatt_amd as sp
sp.SpTensor.synthetic([80, 60, 100], 20000, seed=11).save('/tmp/asan_t.tns')
sp.SpTensor.synthetic([30, 40, 25, 12], 8000, seed=12).save('/tmp/asan_t4.tns')
PY
/tmp/splatt_asan stats  /tmp/asan_t.tns
/tmp/splatt_asan check  /tmp/asan_t.tns
/tmp/splatt_asan convert /tmp/asan_t.tns /tmp/asan_t.bin
/tmp/splatt_asan stats  /tmp/asan_t.bin
/tmp/splatt_asan cpd    /tmp/asan_t.tns  -r 8  -i 10 --nowrite
/tmp/splatt_asan cpd    /tmp/asan_t4.tns -r 12 -i 8  --nowrite
/tmp/splatt_asan bench  /tmp/asan_t.tns  -r 16 -N 2
echo "sanitizer pass clean"
