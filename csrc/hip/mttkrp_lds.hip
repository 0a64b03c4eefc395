// LDS-staged flat MTTKRP (v5) — factor-row tiles staged in LDS.
//
// PMC evidence (profiles/ROUND1_PROFILING.md): the flat kernel is
// TA-bound (~87% TA_TA_BUSY) on cache-resident tensors — every nonzero
// costs ~4 L1 tag lookups for its two factor-row gathers. This kernel
// removes the L1 path for ONE gathered mode entirely: the CSF build
// buckets nonzeros by that mode's row range (gather-tile sort,
// splatt_amd/csf.py), each workgroup's nnz range lies inside one bucket,
// and the bucket's factor-row slice is staged ONCE into LDS by the whole
// block; per-nonzero reads become ds_read (256 B/clk/CU, no tag path).
// This is the north-star design: LDS factor tiles + privatized register
// accumulators + one hardware atomic per output-row run (BASELINE.json).
//
// Work decomposition: host passes per-block descriptors (nnz range +
// bucket row0) built from the bucket boundaries, so the kernel needs no
// searching. Output is at the CSF root (ALLMODE dispatch); other output
// depths fall back to the v2 kernel.
#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstdlib>

#include "store_types.hpp"

namespace {
using splatt_store::bf16;
using splatt_store::VecElemT;
using splatt_store::to_compute;

constexpr int WAVE = 64;
constexpr int WPB = 4;   // waves per block

__device__ __forceinline__ int64_t min64(int64_t a, int64_t b) { return a < b ? a : b; }

template <typename V>
__device__ __forceinline__ void atomic_add_g(V * p, V v) {
  unsafeAtomicAdd(p, v);
}

template <typename T>
__device__ __forceinline__ T ldnt(const T * p) {
  return __builtin_nontemporal_load(p);
}

// i0/m0 = the STAGED (bucketed) level; i1.. = the remaining levels.
template <typename V, int F, int NOTHER>
__global__ void __launch_bounds__(WPB * WAVE)
mttkrp_flat5_kern(const int32_t * __restrict__ key,
                  const int32_t * __restrict__ i0,
                  const int32_t * __restrict__ i1,
                  const int32_t * __restrict__ i2,
                  const int32_t * __restrict__ i3,
                  const V * __restrict__ m0, const V * __restrict__ m1,
                  const V * __restrict__ m2, const V * __restrict__ m3,
                  const V * __restrict__ vals,
                  const int64_t * __restrict__ blk_start,
                  const int64_t * __restrict__ blk_end,
                  const int32_t * __restrict__ blk_row0,
                  int32_t chunk, int32_t dim0,
                  V * __restrict__ out) {
  constexpr int GB = 8;   // 91 VGPR, 5 waves/SIMD (GB=6 at 24 waves/CU measured -5%)
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  V * smem = reinterpret_cast<V *>(smem_raw);

  const int b = blockIdx.x;
  const int64_t b0 = blk_start[b];
  const int64_t b1 = blk_end[b];
  const int32_t row0 = blk_row0[b];
  const int nrows = (int)min64((int64_t)chunk, (int64_t)dim0 - row0);

  // cooperative stage: bucket's m0 rows -> LDS (coalesced, 16B per lane;
  // base is 16B-aligned because F is even for every spec rank)
  {
    constexpr int VEC = 16 / sizeof(V);
    using Vec = __attribute__((ext_vector_type(VEC))) V;
    const int nel = nrows * F;
    const int nvec = nel / VEC;
    const int tid = threadIdx.x;
    const Vec * src = reinterpret_cast<const Vec *>(m0 + (int64_t)row0 * F);
    Vec * dst = reinterpret_cast<Vec *>(smem);
    for (int ve = tid; ve < nvec; ve += WPB * WAVE) dst[ve] = src[ve];
    for (int t = nvec * VEC + tid; t < nel; t += WPB * WAVE)
      smem[t] = m0[(int64_t)row0 * F + t];
  }
  __syncthreads();

  // per-wave sub-spans of [b0, b1)
  constexpr int R = WAVE / F;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wv = threadIdx.x / WAVE;
  const int c = lane % F;
  const int g = lane / F;
  const int gbase = g * F;
  const int nsub = WPB * R;
  const int sub = wv * R + g;
  const int64_t total = b1 - b0;
  const int64_t gsz = (total + nsub - 1) / nsub;
  const int64_t p0 = min64(b1, b0 + sub * gsz);
  const int64_t p1 = min64(b1, p0 + gsz);
  if (p0 >= p1) return;

  int32_t cur = key[p0];
  V acc = (V)0;
  for (int64_t pb = p0; pb < p1; pb += F) {
    const int nb = (int)min64((int64_t)F, p1 - pb);
    const int64_t ps = pb + (c < nb ? c : nb - 1);
    const int32_t kreg = ldnt(&key[ps]);
    const int32_t i0reg = ldnt(&i0[ps]);
    const int32_t i1reg = ldnt(&i1[ps]);
    const int32_t i2reg = (NOTHER > 2) ? ldnt(&i2[ps]) : 0;
    const int32_t i3reg = (NOTHER > 3) ? ldnt(&i3[ps]) : 0;
    const V vreg = ldnt(&vals[ps]);
    for (int ub = 0; ub < nb; ub += GB) {
      const int ne = nb - ub < GB ? nb - ub : GB;
      int32_t kk[GB];
      V vv[GB], a0[GB], a1[GB], a2[GB], a3[GB];
      #pragma unroll
      for (int u = 0; u < GB; ++u) {
        const int src = gbase + (u < ne ? ub + u : ub);
        kk[u] = __shfl(kreg, src, WAVE);
        vv[u] = __shfl(vreg, src, WAVE);
        const int32_t j0 = __shfl(i0reg, src, WAVE);
        const int32_t j1 = __shfl(i1reg, src, WAVE);
        a0[u] = smem[(j0 - row0) * F + c];          // LDS, no L1/TA
        a1[u] = m1[(int64_t)j1 * F + c];
        if (NOTHER > 2) {
          const int32_t j2 = __shfl(i2reg, src, WAVE);
          a2[u] = m2[(int64_t)j2 * F + c];
        }
        if (NOTHER > 3) {
          const int32_t j3 = __shfl(i3reg, src, WAVE);
          a3[u] = m3[(int64_t)j3 * F + c];
        }
      }
      #pragma unroll
      for (int u = 0; u < GB; ++u) {
        if (u >= ne) break;
        V x = vv[u] * a0[u] * a1[u];
        if (NOTHER > 2) x *= a2[u];
        if (NOTHER > 3) x *= a3[u];
        if (kk[u] != cur) {
          atomic_add_g(&out[(int64_t)cur * F + c], acc);
          acc = (V)0;
          cur = kk[u];
        }
        acc += x;
      }
    }
  }
  atomic_add_g(&out[(int64_t)cur * F + c], acc);
}

// ------------------------------------------------- packed-stream v6
// v5 loads key/i0/i1(/i2) as 3-4 separate 4-B stream reads per nonzero
// slot; the per-CU TA address path is the measured binder
// (profiles/ROUND1_PROFILING.md), so v6 packs them into ONE int4 word
// per nonzero (built at CSF-build time, splatt_amd/csf.py): word layout
// x = output key (root label), y = staged-level label, z/w = remaining
// levels. Stream tag lookups per slot drop from 4-5 to 2 (pack + vals).
//
// S = factor STORAGE type (defaults to the compute type V). S=float or
// S=bf16 with V=double is the documented reduced-precision factor-store
// mode for HBM-bound shapes: gathered rows shrink 2-4x in cache lines
// while every multiply-accumulate stays f64 (ROADMAP item 2b).
template <typename V, int F, int NOTHER, typename S = V, int GBP = 8,
          int WPB6K = WPB>
__global__ void __launch_bounds__(WPB6K * WAVE)
mttkrp_flat6_kern(const int * __restrict__ pack_raw,
                  const S * __restrict__ m0, const S * __restrict__ m1,
                  const S * __restrict__ m2,
                  const V * __restrict__ vals,
                  const int64_t * __restrict__ blk_start,
                  const int64_t * __restrict__ blk_end,
                  const int32_t * __restrict__ blk_row0,
                  int32_t chunk, int32_t dim0,
                  V * __restrict__ out) {
  using Pack = __attribute__((ext_vector_type(4))) int;
  const Pack * __restrict__ pack = reinterpret_cast<const Pack *>(pack_raw);
  constexpr int GB = (F >= GBP) ? GBP : F;
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  S * smem = reinterpret_cast<S *>(smem_raw);

  const int b = blockIdx.x;
  const int64_t b0 = blk_start[b];
  const int64_t b1 = blk_end[b];
  const int32_t row0 = blk_row0[b];
  const int nrows = (int)min64((int64_t)chunk, (int64_t)dim0 - row0);

  {
    const int nel = nrows * F;
    const int tid = threadIdx.x;
    if ((((int64_t)row0 * F * sizeof(S)) & 15) == 0) {
      // vectorize the stage copy through an integral proxy of the same
      // width (struct element types cannot form ext_vector_type)
      using E = typename VecElemT<S>::type;
      constexpr int VEC = 16 / sizeof(E);
      using Vec = __attribute__((ext_vector_type(VEC))) E;
      const int nvec = nel / VEC;
      const Vec * src = reinterpret_cast<const Vec *>(m0 + (int64_t)row0 * F);
      Vec * dst = reinterpret_cast<Vec *>(smem);
      for (int ve = tid; ve < nvec; ve += WPB6K * WAVE) dst[ve] = src[ve];
      for (int t = nvec * VEC + tid; t < nel; t += WPB6K * WAVE)
        smem[t] = m0[(int64_t)row0 * F + t];
    } else {
      for (int t = tid; t < nel; t += WPB6K * WAVE)
        smem[t] = m0[(int64_t)row0 * F + t];
    }
  }
  __syncthreads();

  constexpr int R = WAVE / F;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wv = threadIdx.x / WAVE;
  const int c = lane % F;
  const int g = lane / F;
  const int gbase = g * F;
  const int nsub = WPB6K * R;
  const int sub = wv * R + g;
  const int64_t total = b1 - b0;
  const int64_t gsz = (total + nsub - 1) / nsub;
  const int64_t p0 = min64(b1, b0 + sub * gsz);
  const int64_t p1 = min64(b1, p0 + gsz);
  if (p0 >= p1) return;

  int32_t cur = pack[p0].x;
  V acc = (V)0;
  for (int64_t pb = p0; pb < p1; pb += F) {
    const int nb = (int)min64((int64_t)F, p1 - pb);
    const int64_t ps = pb + (c < nb ? c : nb - 1);
    const Pack preg = __builtin_nontemporal_load(&pack[ps]);
    const V vreg = ldnt(&vals[ps]);
    for (int ub = 0; ub < nb; ub += GB) {
      const int ne = nb - ub < GB ? nb - ub : GB;
      int32_t kk[GB];
      V vv[GB];
      S a0[GB], a1[GB], a2[GB];
      #pragma unroll
      for (int u = 0; u < GB; ++u) {
        const int src = gbase + (u < ne ? ub + u : ub);
        kk[u] = __shfl(preg.x, src, WAVE);
        vv[u] = __shfl(vreg, src, WAVE);
        const int32_t j0 = __shfl(preg.y, src, WAVE);
        const int32_t j1 = __shfl(preg.z, src, WAVE);
        a0[u] = smem[(j0 - row0) * F + c];          // LDS, no L1/TA
        a1[u] = m1[(int64_t)j1 * F + c];
        if (NOTHER > 2) {
          const int32_t j2 = __shfl(preg.w, src, WAVE);
          a2[u] = m2[(int64_t)j2 * F + c];
        }
      }
      #pragma unroll
      for (int u = 0; u < GB; ++u) {
        if (u >= ne) break;
        V x = vv[u] * to_compute(a0[u], (V)0) * to_compute(a1[u], (V)0);
        if (NOTHER > 2) x *= to_compute(a2[u], (V)0);
        if (kk[u] != cur) {
          atomic_add_g(&out[(int64_t)cur * F + c], acc);
          acc = (V)0;
          cur = kk[u];
        }
        acc += x;
      }
    }
  }
  atomic_add_g(&out[(int64_t)cur * F + c], acc);
}

template <typename V, typename S = V>
void launch_flat6(const int32_t * pack, const S * const mats[3],
                  const V * vals, const int64_t * blk_start,
                  const int64_t * blk_end, const int32_t * blk_row0,
                  int64_t nblocks, int32_t chunk, int32_t dim0, V * out,
                  int rank, int nother, hipStream_t st) {
  const char * we = getenv("SPLATT_V6_WPB");
  const int wpb6 = (we && atoi(we) == 8) ? 8 : WPB;
  dim3 grid((uint32_t)nblocks), block(wpb6 * WAVE);
  const size_t lds = (size_t)chunk * rank * sizeof(S);
  // gather-batch depth A/B lever (register pressure vs loads in flight)
  const char * ge = getenv("SPLATT_V6_GB");
  const int gb = ge ? atoi(ge) : 8;
#define ARGS6 pack, mats[0], mats[1], mats[2], vals, blk_start, blk_end, \
              blk_row0, chunk, dim0, out
#define L6(F_, N_) \
  if (wpb6 == 8) { \
    hipLaunchKernelGGL((mttkrp_flat6_kern<V, F_, N_, S, 8, 8>), grid, block, lds, st, ARGS6); \
  } else switch (gb) { \
    case 4:  hipLaunchKernelGGL((mttkrp_flat6_kern<V, F_, N_, S, 4>), grid, block, lds, st, ARGS6); break; \
    case 6:  hipLaunchKernelGGL((mttkrp_flat6_kern<V, F_, N_, S, 6>), grid, block, lds, st, ARGS6); break; \
    case 12: hipLaunchKernelGGL((mttkrp_flat6_kern<V, F_, N_, S, 12>), grid, block, lds, st, ARGS6); break; \
    case 16: hipLaunchKernelGGL((mttkrp_flat6_kern<V, F_, N_, S, 16>), grid, block, lds, st, ARGS6); break; \
    default: hipLaunchKernelGGL((mttkrp_flat6_kern<V, F_, N_, S, 8>), grid, block, lds, st, ARGS6); break; }
#define L6F(N_) \
  switch (rank) { case 4: L6(4, N_); break; case 8: L6(8, N_); break; \
                  case 16: L6(16, N_); break; case 32: L6(32, N_); break; \
                  default: L6(64, N_); break; }
  if (nother == 2) { L6F(2); } else { L6F(3); }
#undef L6F
#undef L6
#undef ARGS6
}

template <typename V>
void launch_flat5(const int32_t * key, const int32_t * const idx[4],
                  const V * const mats[4], const V * vals,
                  const int64_t * blk_start, const int64_t * blk_end,
                  const int32_t * blk_row0, int64_t nblocks, int32_t chunk,
                  int32_t dim0, V * out, int rank, int nother,
                  hipStream_t st) {
  dim3 grid((uint32_t)nblocks), block(WPB * WAVE);
  const size_t lds = (size_t)chunk * rank * sizeof(V);
#define ARGS key, idx[0], idx[1], idx[2], idx[3], mats[0], mats[1], mats[2], \
             mats[3], vals, blk_start, blk_end, blk_row0, chunk, dim0, out
#define L5(F_, N_) \
  hipLaunchKernelGGL((mttkrp_flat5_kern<V, F_, N_>), grid, block, lds, st, ARGS)
#define L5F(N_) \
  switch (rank) { case 4: L5(4, N_); break; case 8: L5(8, N_); break; \
                  case 16: L5(16, N_); break; case 32: L5(32, N_); break; \
                  default: L5(64, N_); break; }
  switch (nother) {
    case 2: L5F(2); break;
    case 3: L5F(3); break;
    default: L5F(4); break;
  }
#undef L5F
#undef L5
#undef ARGS
}

}  // namespace

extern "C" void splatt_hip_mttkrp_flat6_f64(
    const int32_t * pack, const double * m0, const double * m1,
    const double * m2, const double * vals, const int64_t * blk_start,
    const int64_t * blk_end, const int32_t * blk_row0, int64_t nblocks,
    int32_t chunk, int32_t dim0, double * out, int rank, int nother,
    void * stream) {
  const double * mats[3] = {m0, m1, m2};
  launch_flat6<double>(pack, mats, vals, blk_start, blk_end, blk_row0,
                       nblocks, chunk, dim0, out, rank, nother,
                       (hipStream_t)stream);
}

extern "C" void splatt_hip_mttkrp_flat6_f32(
    const int32_t * pack, const float * m0, const float * m1,
    const float * m2, const float * vals, const int64_t * blk_start,
    const int64_t * blk_end, const int32_t * blk_row0, int64_t nblocks,
    int32_t chunk, int32_t dim0, float * out, int rank, int nother,
    void * stream) {
  const float * mats[3] = {m0, m1, m2};
  launch_flat6<float>(pack, mats, vals, blk_start, blk_end, blk_row0,
                      nblocks, chunk, dim0, out, rank, nother,
                      (hipStream_t)stream);
}

// reduced-precision factor STORAGE (f64 accumulation) — ROADMAP 2b
extern "C" void splatt_hip_mttkrp_flat6_f64f32(
    const int32_t * pack, const float * m0, const float * m1,
    const float * m2, const double * vals, const int64_t * blk_start,
    const int64_t * blk_end, const int32_t * blk_row0, int64_t nblocks,
    int32_t chunk, int32_t dim0, double * out, int rank, int nother,
    void * stream) {
  const float * mats[3] = {m0, m1, m2};
  launch_flat6<double, float>(pack, mats, vals, blk_start, blk_end,
                              blk_row0, nblocks, chunk, dim0, out, rank,
                              nother, (hipStream_t)stream);
}

extern "C" void splatt_hip_mttkrp_flat6_f64bf16(
    const int32_t * pack, const uint16_t * m0, const uint16_t * m1,
    const uint16_t * m2, const double * vals, const int64_t * blk_start,
    const int64_t * blk_end, const int32_t * blk_row0, int64_t nblocks,
    int32_t chunk, int32_t dim0, double * out, int rank, int nother,
    void * stream) {
  const bf16 * mats[3] = {(const bf16*)m0, (const bf16*)m1,
                          (const bf16*)m2};
  launch_flat6<double, bf16>(pack, mats, vals, blk_start, blk_end,
                             blk_row0, nblocks, chunk, dim0, out, rank,
                             nother, (hipStream_t)stream);
}

extern "C" void splatt_hip_mttkrp_flat5_f64(
    const int32_t * key, const int32_t * i0, const int32_t * i1,
    const int32_t * i2, const int32_t * i3, const double * m0,
    const double * m1, const double * m2, const double * m3,
    const double * vals, const int64_t * blk_start, const int64_t * blk_end,
    const int32_t * blk_row0, int64_t nblocks, int32_t chunk, int32_t dim0,
    double * out, int rank, int nother, void * stream) {
  const int32_t * idx[4] = {i0, i1, i2, i3};
  const double * mats[4] = {m0, m1, m2, m3};
  launch_flat5<double>(key, idx, mats, vals, blk_start, blk_end, blk_row0,
                       nblocks, chunk, dim0, out, rank, nother,
                       (hipStream_t)stream);
}

extern "C" void splatt_hip_mttkrp_flat5_f32(
    const int32_t * key, const int32_t * i0, const int32_t * i1,
    const int32_t * i2, const int32_t * i3, const float * m0,
    const float * m1, const float * m2, const float * m3,
    const float * vals, const int64_t * blk_start, const int64_t * blk_end,
    const int32_t * blk_row0, int64_t nblocks, int32_t chunk, int32_t dim0,
    float * out, int rank, int nother, void * stream) {
  const int32_t * idx[4] = {i0, i1, i2, i3};
  const float * mats[4] = {m0, m1, m2, m3};
  launch_flat5<float>(key, idx, mats, vals, blk_start, blk_end, blk_row0,
                      nblocks, chunk, dim0, out, rank, nother,
                      (hipStream_t)stream);
}
