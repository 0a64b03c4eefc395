// CDNA4 (gfx950) MTTKRP kernels over flat CSF — the flagship hot path.
//
// Capability parity: the root/internal/leaf 3-mode kernel family of the
// reference (src/mttkrp.c:390-665 p_csf_mttkrp_{root3,intl3,leaf3}_*).
// Fresh MI355X design, not a port:
//   * work decomposition = equal NNZ SPANS per 64-lane wavefront (perfect
//     load balance on power-law fibers), not slice/tile CCP partitions;
//     each wave binary-searches its span's (fiber, slice) start once and
//     then walks the CSF levels forward.
//   * lane layout: 64 lanes = F columns x R=64/F nnz sub-slots; fiber
//     accumulation is a register butterfly (__shfl_xor) across sub-slots,
//     replacing the reference's thread-private accumF buffers.
//   * cross-wave output conflicts resolved by hardware f64/f32 atomic adds
//     at slice/fiber granularity (CDNA4-native global_atomic_add_f64),
//     replacing the reference's 1024-entry mutex pool (mutex_pool.c).
// Compile: hipcc --offload-arch=gfx950 -O3 (see setup.py).
#include <hip/hip_runtime.h>
#include <cstdint>

namespace {

constexpr int WAVE = 64;

__device__ __forceinline__ int64_t min64(int64_t a, int64_t b) { return a < b ? a : b; }

template <typename V>
__device__ __forceinline__ void atomic_add_g(V * p, V v) {
  // CDNA has native global fadd for f32/f64; unsafeAtomicAdd emits it
  // (plain atomicAdd lowers to a CAS loop without -munsafe-fp-atomics).
  unsafeAtomicAdd(p, v);
}

// first index i in [0,n) with a[i] > key
__device__ __forceinline__ int64_t upper_bound_i64(
    const int64_t * __restrict__ a, int64_t n, int64_t key) {
  int64_t lo = 0, hi = n;
  while (lo < hi) {
    const int64_t mid = (lo + hi) >> 1;
    if (a[mid] <= key) lo = mid + 1; else hi = mid;
  }
  return lo;
}

// butterfly-reduce `x` across the R = 64/F nnz sub-slots of a wave; every
// lane ends with the full sum for its column.
template <typename V, int F>
__device__ __forceinline__ V subslot_reduce(V x) {
  #pragma unroll
  for (int off = F; off < WAVE; off <<= 1) x += __shfl_xor(x, off, WAVE);
  return x;
}

// ------------------------------------------------------------------ root3
// out[slice_row, :] += sum_fibers A1[fid1,:] (.) (sum_nnz v * A2[fid2,:])
template <typename V, int F>
__global__ void __launch_bounds__(256)
mttkrp_root3_kern(const int64_t * __restrict__ fptr0,
                  const int32_t * __restrict__ fids0,
                  const int64_t * __restrict__ fptr1,
                  const int32_t * __restrict__ fids1,
                  const int32_t * __restrict__ fids2,
                  const V * __restrict__ vals,
                  int64_t nslices, int64_t nfibs, int64_t nnz,
                  int64_t span,
                  const V * __restrict__ A1, const V * __restrict__ A2,
                  V * __restrict__ out) {
  constexpr int R = WAVE / F;
  const int lane = threadIdx.x & (WAVE - 1);
  const int64_t wid = (int64_t)blockIdx.x * (blockDim.x / WAVE)
                      + (threadIdx.x / WAVE);
  const int c = lane % F;
  const int r = lane / F;
  const int64_t j0 = wid * span;
  if (j0 >= nnz) return;
  const int64_t j1 = min64(nnz, j0 + span);

  int64_t f = upper_bound_i64(fptr1, nfibs + 1, j0) - 1;
  int64_t s = upper_bound_i64(fptr0, nslices + 1, f) - 1;
  int64_t slice_end_fiber = fptr0[s + 1];

  V slice_acc = (V)0;
  int64_t j = j0;
  while (j < j1) {
    const int64_t fiber_end = fptr1[f + 1];
    const int64_t jend = min64(fiber_end, j1);
    V acc = (V)0;
    for (int64_t p = j + r; p < jend; p += R)
      acc += vals[p] * A2[(int64_t)fids2[p] * F + c];
    acc = subslot_reduce<V, F>(acc);
    slice_acc += acc * A1[(int64_t)fids1[f] * F + c];
    j = jend;
    if (j == fiber_end) {
      ++f;
      if (f == slice_end_fiber && j < j1) {
        if (r == 0) {
          const int64_t orow = fids0 ? (int64_t)fids0[s] : s;
          atomic_add_g(&out[orow * F + c], slice_acc);
        }
        slice_acc = (V)0;
        ++s;
        slice_end_fiber = fptr0[s + 1];
      }
    }
  }
  if (r == 0) {
    const int64_t orow = fids0 ? (int64_t)fids0[s] : s;
    atomic_add_g(&out[orow * F + c], slice_acc);
  }
}

// ------------------------------------------------------------------ intl3
// out[fid1, :] += A0[slice_row,:] (.) (sum_nnz v * A2[fid2,:])
template <typename V, int F>
__global__ void __launch_bounds__(256)
mttkrp_intl3_kern(const int64_t * __restrict__ fptr0,
                  const int32_t * __restrict__ fids0,
                  const int64_t * __restrict__ fptr1,
                  const int32_t * __restrict__ fids1,
                  const int32_t * __restrict__ fids2,
                  const V * __restrict__ vals,
                  int64_t nslices, int64_t nfibs, int64_t nnz,
                  int64_t span,
                  const V * __restrict__ A0, const V * __restrict__ A2,
                  V * __restrict__ out) {
  constexpr int R = WAVE / F;
  const int lane = threadIdx.x & (WAVE - 1);
  const int64_t wid = (int64_t)blockIdx.x * (blockDim.x / WAVE)
                      + (threadIdx.x / WAVE);
  const int c = lane % F;
  const int r = lane / F;
  const int64_t j0 = wid * span;
  if (j0 >= nnz) return;
  const int64_t j1 = min64(nnz, j0 + span);

  int64_t f = upper_bound_i64(fptr1, nfibs + 1, j0) - 1;
  int64_t s = upper_bound_i64(fptr0, nslices + 1, f) - 1;
  int64_t slice_end_fiber = fptr0[s + 1];
  V arow = A0[(fids0 ? (int64_t)fids0[s] : s) * F + c];

  int64_t j = j0;
  while (j < j1) {
    const int64_t fiber_end = fptr1[f + 1];
    const int64_t jend = min64(fiber_end, j1);
    V acc = (V)0;
    for (int64_t p = j + r; p < jend; p += R)
      acc += vals[p] * A2[(int64_t)fids2[p] * F + c];
    acc = subslot_reduce<V, F>(acc);
    if (r == 0)
      atomic_add_g(&out[(int64_t)fids1[f] * F + c], acc * arow);
    j = jend;
    if (j == fiber_end) {
      ++f;
      if (f == slice_end_fiber && j < j1) {
        ++s;
        slice_end_fiber = fptr0[s + 1];
        arow = A0[(fids0 ? (int64_t)fids0[s] : s) * F + c];
      }
    }
  }
}

// ------------------------------------------------------------------ leaf3
// out[fid2, :] += v * (A0[slice_row,:] (.) A1[fid1,:])   (scatter-heavy)
template <typename V, int F>
__global__ void __launch_bounds__(256)
mttkrp_leaf3_kern(const int64_t * __restrict__ fptr0,
                  const int32_t * __restrict__ fids0,
                  const int64_t * __restrict__ fptr1,
                  const int32_t * __restrict__ fids1,
                  const int32_t * __restrict__ fids2,
                  const V * __restrict__ vals,
                  int64_t nslices, int64_t nfibs, int64_t nnz,
                  int64_t span,
                  const V * __restrict__ A0, const V * __restrict__ A1,
                  V * __restrict__ out) {
  constexpr int R = WAVE / F;
  const int lane = threadIdx.x & (WAVE - 1);
  const int64_t wid = (int64_t)blockIdx.x * (blockDim.x / WAVE)
                      + (threadIdx.x / WAVE);
  const int c = lane % F;
  const int r = lane / F;
  const int64_t j0 = wid * span;
  if (j0 >= nnz) return;
  const int64_t j1 = min64(nnz, j0 + span);

  int64_t f = upper_bound_i64(fptr1, nfibs + 1, j0) - 1;
  int64_t s = upper_bound_i64(fptr0, nslices + 1, f) - 1;
  int64_t slice_end_fiber = fptr0[s + 1];
  V arow = A0[(fids0 ? (int64_t)fids0[s] : s) * F + c];

  int64_t j = j0;
  while (j < j1) {
    const int64_t fiber_end = fptr1[f + 1];
    const int64_t jend = min64(fiber_end, j1);
    const V w = arow * A1[(int64_t)fids1[f] * F + c];
    for (int64_t p = j + r; p < jend; p += R)
      atomic_add_g(&out[(int64_t)fids2[p] * F + c], vals[p] * w);
    j = jend;
    if (j == fiber_end) {
      ++f;
      if (f == slice_end_fiber && j < j1) {
        ++s;
        slice_end_fiber = fptr0[s + 1];
        arow = A0[(fids0 ? (int64_t)fids0[s] : s) * F + c];
      }
    }
  }
}

// ------------------------------------------- generic-rank fallback kernels
// one wave per fiber (grid-stride); columns chunked by 64. Correctness path
// for ranks outside {4,8,16,32,64}.
template <typename V, int WHICH>  // 0 root, 1 intl, 2 leaf
__global__ void __launch_bounds__(256)
mttkrp3_generic_kern(const int64_t * __restrict__ fptr0,
                     const int32_t * __restrict__ fids0,
                     const int64_t * __restrict__ fptr1,
                     const int32_t * __restrict__ fids1,
                     const int32_t * __restrict__ fids2,
                     const V * __restrict__ vals,
                     int64_t nslices, int64_t nfibs, int64_t nnz, int rank,
                     const V * __restrict__ Ma, const V * __restrict__ Mb,
                     V * __restrict__ out) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int64_t wid0 = (int64_t)blockIdx.x * (blockDim.x / WAVE)
                       + (threadIdx.x / WAVE);
  const int64_t nwaves = (int64_t)gridDim.x * (blockDim.x / WAVE);
  for (int64_t f = wid0; f < nfibs; f += nwaves) {
    const int64_t s = upper_bound_i64(fptr0, nslices + 1, f) - 1;
    const int64_t srow = fids0 ? (int64_t)fids0[s] : s;
    const int64_t jbeg = fptr1[f], jend = fptr1[f + 1];
    for (int cb = 0; cb < rank; cb += WAVE) {
      const int c = cb + lane;
      if (c >= rank) break;
      if (WHICH == 2) {
        const V w = Ma[srow * rank + c] * Mb[(int64_t)fids1[f] * rank + c];
        for (int64_t p = jbeg; p < jend; ++p)
          atomic_add_g(&out[(int64_t)fids2[p] * rank + c], vals[p] * w);
      } else {
        V acc = (V)0;
        for (int64_t p = jbeg; p < jend; ++p)
          acc += vals[p] * Mb[(int64_t)fids2[p] * rank + c];
        if (WHICH == 0) {
          acc *= Ma[(int64_t)fids1[f] * rank + c];  // Ma = A1 here
          atomic_add_g(&out[srow * rank + c], acc);
        } else {
          acc *= Ma[srow * rank + c];               // Ma = A0 here
          atomic_add_g(&out[(int64_t)fids1[f] * rank + c], acc);
        }
      }
    }
  }
}

// --------------------------------------------------------------- launchers

inline int64_t pick_span(int64_t nnz) {
  // enough waves to fill 256 CUs x 8 XCDs many times over, but >=256 nnz
  // of work per wave so the binary searches amortize
  int64_t span = nnz / 65536;
  if (span < 256) span = 256;
  if (span > 16384) span = 16384;
  return span;
}

// dispatch a spec kernel over the supported rank set; KERN is a function
// template name, so this is a macro (template-template params only bind
// class templates).
#define LAUNCH_SPEC(KERN, V, F_RT)                                            \
  do {                                                                        \
    const int64_t span_ = pick_span(nnz);                                     \
    const int64_t nwaves_ = (nnz + span_ - 1) / span_;                        \
    const int wpb_ = 4; /* 256-thread blocks */                               \
    const int64_t nblocks_ = (nwaves_ + wpb_ - 1) / wpb_;                     \
    dim3 grid_((uint32_t)nblocks_), block_(wpb_ * WAVE);                      \
    switch (F_RT) {                                                           \
      case 4:  hipLaunchKernelGGL((KERN<V, 4>),  grid_, block_, 0, st, fptr0, fids0, fptr1, fids1, fids2, vals, nslices, nfibs, nnz, span_, Ma, Mb, out); break; \
      case 8:  hipLaunchKernelGGL((KERN<V, 8>),  grid_, block_, 0, st, fptr0, fids0, fptr1, fids1, fids2, vals, nslices, nfibs, nnz, span_, Ma, Mb, out); break; \
      case 16: hipLaunchKernelGGL((KERN<V, 16>), grid_, block_, 0, st, fptr0, fids0, fptr1, fids1, fids2, vals, nslices, nfibs, nnz, span_, Ma, Mb, out); break; \
      case 32: hipLaunchKernelGGL((KERN<V, 32>), grid_, block_, 0, st, fptr0, fids0, fptr1, fids1, fids2, vals, nslices, nfibs, nnz, span_, Ma, Mb, out); break; \
      default: hipLaunchKernelGGL((KERN<V, 64>), grid_, block_, 0, st, fptr0, fids0, fptr1, fids1, fids2, vals, nslices, nfibs, nnz, span_, Ma, Mb, out); break; \
    }                                                                         \
  } while (0)

template <typename V>
void launch_generic(int which, int rank, const int64_t * fptr0,
                    const int32_t * fids0, const int64_t * fptr1,
                    const int32_t * fids1, const int32_t * fids2,
                    const V * vals, int64_t nslices, int64_t nfibs,
                    int64_t nnz, const V * Ma, const V * Mb, V * out,
                    hipStream_t st) {
  const int wpb = 4;
  int64_t nblocks = (nfibs + wpb - 1) / wpb;
  if (nblocks > 16384) nblocks = 16384;
  dim3 grid((uint32_t)nblocks), block(wpb * WAVE);
  if (which == 0)
    hipLaunchKernelGGL((mttkrp3_generic_kern<V, 0>), grid, block, 0, st, fptr0, fids0, fptr1, fids1, fids2, vals, nslices, nfibs, nnz, rank, Ma, Mb, out);
  else if (which == 1)
    hipLaunchKernelGGL((mttkrp3_generic_kern<V, 1>), grid, block, 0, st, fptr0, fids0, fptr1, fids1, fids2, vals, nslices, nfibs, nnz, rank, Ma, Mb, out);
  else
    hipLaunchKernelGGL((mttkrp3_generic_kern<V, 2>), grid, block, 0, st, fptr0, fids0, fptr1, fids1, fids2, vals, nslices, nfibs, nnz, rank, Ma, Mb, out);
}

inline bool spec_ok(int F) {
  return F == 4 || F == 8 || F == 16 || F == 32 || F == 64;
}

}  // namespace

#define DEFINE_ENTRY(NAME, VTYPE, KERN, WHICH)                               \
  extern "C" void NAME(const int64_t * fptr0, const int32_t * fids0,         \
                       const int64_t * fptr1, const int32_t * fids1,         \
                       const int32_t * fids2, const VTYPE * vals,            \
                       int64_t nslices, int64_t nfibs, int64_t nnz,          \
                       const VTYPE * Ma, const VTYPE * Mb, VTYPE * out,      \
                       int rank, void * stream) {                            \
    hipStream_t st = (hipStream_t)stream;                                    \
    if (spec_ok(rank))                                                       \
      LAUNCH_SPEC(KERN, VTYPE, rank);                                        \
    else                                                                     \
      launch_generic<VTYPE>(WHICH, rank, fptr0, fids0, fptr1, fids1, fids2,  \
                            vals, nslices, nfibs, nnz, Ma, Mb, out, st);     \
  }

DEFINE_ENTRY(splatt_hip_mttkrp_root3_f64, double, mttkrp_root3_kern, 0)
DEFINE_ENTRY(splatt_hip_mttkrp_root3_f32, float,  mttkrp_root3_kern, 0)
DEFINE_ENTRY(splatt_hip_mttkrp_intl3_f64, double, mttkrp_intl3_kern, 1)
DEFINE_ENTRY(splatt_hip_mttkrp_intl3_f32, float,  mttkrp_intl3_kern, 1)
DEFINE_ENTRY(splatt_hip_mttkrp_leaf3_f64, double, mttkrp_leaf3_kern, 2)
DEFINE_ENTRY(splatt_hip_mttkrp_leaf3_f32, float,  mttkrp_leaf3_kern, 2)

extern "C" int splatt_hip_kernels_arch(void) { return 950; }
