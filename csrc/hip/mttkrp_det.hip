// Deterministic flat MTTKRP (gfx950) — bitwise-reproducible GPU path.
//
// The default flat kernels (mttkrp_flat.hip) emit one hardware atomic add
// per output-key run; atomic arrival order varies between runs, so results
// differ in the last ulp. The reference's answer to reproducibility is its
// serial mttkrp_stream oracle (reference src/mttkrp.c stream path); this
// file gives the same property AT DEVICE SPEED for depth-0 (root-sorted)
// streams, i.e. the default ALLMODE policy where every mode's stream is
// globally sorted by output key:
//
//   * each walker (a wavefront column group with its own contiguous nnz
//     span, exactly the flat-v2 decomposition) folds key runs in registers
//     as before, but emits with PLAIN STORES:
//       - a key strictly inside the span is exclusive to this walker
//         (keys are sorted-contiguous), so its run sum is final;
//       - the first/last key of the span may be shared with neighbouring
//         walkers, so their partials go to a per-walker side buffer
//         (2 slots x rank) instead of `out`.
//   * a fixup kernel combines side-buffer partials IN WALKER ORDER: the
//     unique first contributor of each boundary key claims it (detected
//     from the stream: p0 == 0 or key[p0-1] != key[p0]) and scans forward
//     over the (almost always 1-2) walkers sharing it.
//
// Everything is a pure function of the (deterministically built) stream,
// so two runs produce bitwise-identical outputs. Requires depth-0 streams
// and a spec rank (4/8/16/32/64, == the column-group width F); the Python
// dispatcher enforces both and says why when they don't hold.
#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdlib>

#include "store_types.hpp"

namespace {

constexpr int WAVE = 64;

__device__ __forceinline__ int64_t min64(int64_t a, int64_t b) {
  return a < b ? a : b;
}

template <typename T>
__device__ __forceinline__ T ldnt(const T * p) {
  return __builtin_nontemporal_load(p);
}

// walker -> its contiguous nnz range (the flat-v2 span decomposition:
// wave `wid` covers [wid*span, ..), split into R = WAVE/F group spans)
__device__ __forceinline__ bool walker_range(int64_t w, int64_t nnz,
                                             int64_t span, int R,
                                             int64_t & p0, int64_t & p1) {
  const int64_t wid = w / R;
  const int g = (int)(w % R);
  const int64_t w0 = wid * span;
  if (w0 >= nnz) return false;
  const int64_t w1 = min64(nnz, w0 + span);
  const int64_t gsz = (w1 - w0 + R - 1) / R;
  p0 = min64(w1, w0 + (int64_t)g * gsz);
  p1 = min64(w1, p0 + gsz);
  return p0 < p1;
}

template <typename V, int F, int NOTHER, int GBP = 8>
__global__ void __launch_bounds__(256)
mttkrp_det_kern(const int32_t * __restrict__ key,
                const int32_t * __restrict__ i0,
                const int32_t * __restrict__ i1,
                const int32_t * __restrict__ i2,
                const int32_t * __restrict__ i3,
                const V * __restrict__ m0, const V * __restrict__ m1,
                const V * __restrict__ m2, const V * __restrict__ m3,
                const V * __restrict__ vals, int64_t nnz, int64_t span,
                V * __restrict__ out, V * __restrict__ side) {
  constexpr int R = WAVE / F;
  constexpr int GB = (F >= GBP) ? GBP : F;
  const int lane = threadIdx.x & (WAVE - 1);
  const int64_t wid = (int64_t)blockIdx.x * (blockDim.x / WAVE)
                      + (threadIdx.x / WAVE);
  const int c = lane % F;
  const int g = lane / F;
  const int gbase = g * F;
  const int64_t w0 = wid * span;
  if (w0 >= nnz) return;
  const int64_t w1 = min64(nnz, w0 + span);
  const int64_t gsz = (w1 - w0 + R - 1) / R;
  const int64_t p0 = min64(w1, w0 + g * gsz);
  const int64_t p1 = min64(w1, p0 + gsz);
  if (p0 >= p1) return;

  const int64_t wkr = wid * R + g;          // global walker id
  const int32_t kf = key[p0];
  const int32_t kl = key[p1 - 1];

  int32_t cur = kf;
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
        a0[u] = m0[(int64_t)j0 * F + c];
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
          // interior keys are exclusive to this walker: plain final store
          if (cur == kf) side[(wkr * 2 + 0) * F + c] = acc;
          else out[(int64_t)cur * F + c] = acc;
          acc = (V)0;
          cur = kk[u];
        }
        acc += x;
      }
    }
  }
  if (cur == kf) side[(wkr * 2 + 0) * F + c] = acc;       // single-key span
  else if (cur == kl) side[(wkr * 2 + 1) * F + c] = acc;
  else out[(int64_t)cur * F + c] = acc;
}

// One F-lane task per walker: claim each boundary key whose first
// contributor this walker is, scan forward over the walkers sharing it
// (in walker order -> deterministic), and write the total.
template <typename V, int F>
__global__ void __launch_bounds__(256)
mttkrp_det_fixup_kern(const int32_t * __restrict__ key, int64_t nnz,
                      int64_t span, int64_t nwalkers,
                      const V * __restrict__ side, V * __restrict__ out) {
  constexpr int R = WAVE / F;
  const int tasks_pb = blockDim.x / F;
  const int64_t w = (int64_t)blockIdx.x * tasks_pb + threadIdx.x / F;
  const int c = threadIdx.x % F;
  if (w >= nwalkers) return;
  int64_t p0, p1;
  if (!walker_range(w, nnz, span, R, p0, p1)) return;
  const int32_t kf = key[p0];
  const int32_t kl = key[p1 - 1];

  auto resolve = [&](int32_t k) {
    V tot = (V)0;
    for (int64_t x = w; x < nwalkers; ++x) {
      int64_t q0, q1;
      if (!walker_range(x, nnz, span, R, q0, q1)) continue;
      const int32_t xf = key[q0];
      if (xf > k) break;
      const int32_t xl = key[q1 - 1];
      if (xf == k) tot += side[(x * 2 + 0) * F + c];
      if (xl == k) tot += side[(x * 2 + 1) * F + c];
      if (xl > k) break;
    }
    out[(int64_t)k * F + c] = tot;
  };

  if (p0 == 0 || key[p0 - 1] != kf) resolve(kf);   // unique first contributor
  if (kl != kf) resolve(kl);   // a key starting strictly inside this span
}

// --------------------------- LDS-staged deterministic kernel (det6)
// The baseline det kernel above runs on the key-sorted stream WITHOUT
// LDS staging (the staged stream is bucket-major), costing ~40% vs the
// default path (round-1 measurement). det6 closes that: it runs on the
// SAME packed bucket-major stream as the default v6 kernel, with output
// privatized PER BUCKET (outb[bucket][row][F]) so that, inside one
// bucket, walker-interior key runs are exclusive and can use plain
// stores; walker-boundary partials go to a side buffer resolved by an
// ordered fixup; a final fold sums the bucket slices in ascending bucket
// order. Every step is a pure function of the stream -> bitwise
// reproducible, at v6's memory behavior. Costs nbuckets*dim0*F*8 bytes
// of workspace (Python gates on SPLATT_DET_MB, falling back to the
// key-sorted kernel above).
constexpr int WPB6 = 4;
inline bool det_spec_ok(int F);

template <typename V, int F, int NOTHER, typename S = V>
__global__ void __launch_bounds__(WPB6 * WAVE)
mttkrp_det6_kern(const int * __restrict__ pack_raw,
                 const S * __restrict__ m0, const S * __restrict__ m1,
                 const S * __restrict__ m2,
                 const V * __restrict__ vals,
                 const int64_t * __restrict__ blk_start,
                 const int64_t * __restrict__ blk_end,
                 const int32_t * __restrict__ blk_row0,
                 int32_t chunk, int32_t dim0, int64_t nrows_out,
                 V * __restrict__ outb, V * __restrict__ side) {
  using Pack = __attribute__((ext_vector_type(4))) int;
  const Pack * __restrict__ pack = reinterpret_cast<const Pack *>(pack_raw);
  constexpr int GB = 8;
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  S * smem = reinterpret_cast<S *>(smem_raw);

  const int b = blockIdx.x;
  const int64_t b0 = blk_start[b];
  const int64_t b1 = blk_end[b];
  const int32_t row0 = blk_row0[b];
  const int nrows = (int)(((int64_t)chunk < (int64_t)dim0 - row0)
                          ? chunk : (int64_t)dim0 - row0);
  {
    const int nel = nrows * F;
    const int tid = threadIdx.x;
    for (int t = tid; t < nel; t += WPB6 * WAVE)
      smem[t] = m0[(int64_t)row0 * F + t];
  }
  __syncthreads();

  constexpr int R = WAVE / F;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wv = threadIdx.x / WAVE;
  const int c = lane % F;
  const int g = lane / F;
  const int gbase = g * F;
  constexpr int NSUB = WPB6 * R;
  const int sub = wv * R + g;
  const int64_t total = b1 - b0;
  const int64_t gsz = (total + NSUB - 1) / NSUB;
  const int64_t p0 = (b0 + sub * gsz < b1) ? b0 + sub * gsz : b1;
  const int64_t p1 = (p0 + gsz < b1) ? p0 + gsz : b1;
  if (p0 >= p1) return;

  const int64_t wkr = (int64_t)b * NSUB + sub;
  const int32_t bucket = row0 / chunk;
  V * __restrict__ myout = outb + (int64_t)bucket * nrows_out * F;
  const int32_t kf = pack[p0].x;
  const int32_t kl = pack[p1 - 1].x;

  int32_t cur = kf;
  V acc = (V)0;
  for (int64_t pb = p0; pb < p1; pb += F) {
    const int nb = (int)((pb + F < p1 ? F : p1 - pb));
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
        a0[u] = smem[(j0 - row0) * F + c];
        a1[u] = m1[(int64_t)j1 * F + c];
        if (NOTHER > 2) {
          const int32_t j2 = __shfl(preg.w, src, WAVE);
          a2[u] = m2[(int64_t)j2 * F + c];
        }
      }
      #pragma unroll
      for (int u = 0; u < GB; ++u) {
        if (u >= ne) break;
        V x = vv[u] * splatt_store::to_compute(a0[u], (V)0)
                    * splatt_store::to_compute(a1[u], (V)0);
        if (NOTHER > 2) x *= splatt_store::to_compute(a2[u], (V)0);
        if (kk[u] != cur) {
          if (cur == kf) side[(wkr * 2 + 0) * F + c] = acc;
          else myout[(int64_t)cur * F + c] = acc;
          acc = (V)0;
          cur = kk[u];
        }
        acc += x;
      }
    }
  }
  if (cur == kf) side[(wkr * 2 + 0) * F + c] = acc;
  else if (cur == kl) side[(wkr * 2 + 1) * F + c] = acc;
  else myout[(int64_t)cur * F + c] = acc;
}

// ordered fixup over block-derived walkers: the unique first contributor
// of each boundary key (within its bucket) scans forward in walker order
template <typename V, int F>
__global__ void __launch_bounds__(256)
mttkrp_det6_fixup_kern(const int * __restrict__ pack_raw,
                       const int64_t * __restrict__ blk_start,
                       const int64_t * __restrict__ blk_end,
                       const int32_t * __restrict__ blk_row0,
                       const int64_t * __restrict__ blk_bucket_p0,
                       int64_t nblocks, int32_t chunk, int64_t nrows_out,
                       const V * __restrict__ side, V * __restrict__ outb) {
  using Pack = __attribute__((ext_vector_type(4))) int;
  const Pack * __restrict__ pack = reinterpret_cast<const Pack *>(pack_raw);
  constexpr int R = WAVE / F;
  constexpr int NSUB = 4 * R;   // == WPB6 * R
  const int tasks_pb = blockDim.x / F;
  const int64_t w = (int64_t)blockIdx.x * tasks_pb + threadIdx.x / F;
  const int c = threadIdx.x % F;
  const int64_t nwalkers = nblocks * NSUB;
  if (w >= nwalkers) return;

  auto range_of = [&](int64_t x, int64_t & q0, int64_t & q1,
                      int32_t & brow0) -> bool {
    const int64_t blk = x / NSUB;
    const int s = (int)(x % NSUB);
    const int64_t a0 = blk_start[blk];
    const int64_t a1 = blk_end[blk];
    brow0 = blk_row0[blk];
    const int64_t gsz = (a1 - a0 + NSUB - 1) / NSUB;
    q0 = (a0 + s * gsz < a1) ? a0 + s * gsz : a1;
    q1 = (q0 + gsz < a1) ? q0 + gsz : a1;
    return q0 < q1;
  };

  int64_t p0, p1;
  int32_t myrow0;
  if (!range_of(w, p0, p1, myrow0)) return;
  const int32_t bucket = myrow0 / chunk;
  const int32_t kf = pack[p0].x;
  const int32_t kl = pack[p1 - 1].x;
  V * __restrict__ myout = outb + (int64_t)bucket * nrows_out * F;

  auto resolve = [&](int32_t k) {
    V tot = (V)0;
    for (int64_t x = w; x < nwalkers; ++x) {
      int64_t q0, q1;
      int32_t brow0;
      if (!range_of(x, q0, q1, brow0)) continue;
      if (brow0 != myrow0) break;                 // left the bucket
      const int32_t xf = pack[q0].x;
      if (xf > k) break;
      const int32_t xl = pack[q1 - 1].x;
      if (xf == k) tot += side[(x * 2 + 0) * F + c];
      if (xl == k) tot += side[(x * 2 + 1) * F + c];
      if (xl > k) break;
    }
    myout[(int64_t)k * F + c] = tot;
  };

  // unique first contributor of kf within this BUCKET: either p0 is the
  // bucket's first stream position (pack[p0-1] belongs to another
  // bucket and may coincidentally equal kf), or the previous element has
  // a different key. The bucket's first position travels with the block
  // table (host-precomputed, splatt_amd/mttkrp.py _stage_blocks).
  const bool is_first = (p0 == blk_bucket_p0[w / NSUB])
                        || (pack[p0 - 1].x != kf);
  if (is_first) resolve(kf);
  if (kl != kf) resolve(kl);
}

// fold bucket slices in ascending bucket order (deterministic; the
// 4-deep bucket unroll keeps enough loads in flight per thread — the
// element count alone is too small to cover DRAM latency)
template <typename V>
__global__ void __launch_bounds__(256)
det6_fold_kern(const V * __restrict__ outb, int64_t nbuckets,
               int64_t elems, V * __restrict__ out) {
  const int64_t t0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t e = t0; e < elems; e += stride) {
    V s0 = (V)0, s1 = (V)0, s2 = (V)0, s3 = (V)0;
    int64_t b = 0;
    for (; b + 4 <= nbuckets; b += 4) {
      s0 += outb[b * elems + e];
      s1 += outb[(b + 1) * elems + e];
      s2 += outb[(b + 2) * elems + e];
      s3 += outb[(b + 3) * elems + e];
    }
    for (; b < nbuckets; ++b) s0 += outb[b * elems + e];
    out[e] = ((s0 + s1) + (s2 + s3));
  }
}

template <typename V>
int launch_det6(const int32_t * pack, const V * const mats[3],
                const V * vals, const int64_t * blk_start,
                const int64_t * blk_end, const int32_t * blk_row0,
                const int64_t * blk_bucket_p0,
                int64_t nblocks, int32_t chunk, int32_t dim0,
                int64_t nrows_out, int64_t nbuckets, V * outb, V * side,
                V * out, int rank, int nother, hipStream_t st) {
  if (!det_spec_ok(rank) || nother > 3) return -1;
  (void)hipMemsetAsync(outb, 0,
                       (size_t)nbuckets * nrows_out * rank * sizeof(V), st);
  constexpr int WAVE_ = 64;
  // single-key walkers never write their slot-1 side entry: zero both
  const int64_t nwalk = nblocks * 4 * (WAVE_ / rank);
  (void)hipMemsetAsync(side, 0, (size_t)nwalk * 2 * rank * sizeof(V), st);
  dim3 grid((uint32_t)nblocks), block(WPB6 * WAVE_);
  const size_t lds = (size_t)chunk * rank * sizeof(V);
#define D6(F_, N_) \
  { hipLaunchKernelGGL((mttkrp_det6_kern<V, F_, N_>), grid, block, lds, st, \
        pack, mats[0], mats[1], mats[2], vals, blk_start, blk_end, \
        blk_row0, chunk, dim0, nrows_out, outb, side); \
    const int tpb = 256; \
    const int64_t nw = nblocks * (4 * (WAVE_ / F_)); \
    const int64_t fb = (nw + tpb / F_ - 1) / (tpb / F_); \
    hipLaunchKernelGGL((mttkrp_det6_fixup_kern<V, F_>), \
        dim3((uint32_t)fb), dim3(tpb), 0, st, pack, blk_start, blk_end, \
        blk_row0, blk_bucket_p0, nblocks, chunk, nrows_out, side, outb); }
#define D6F(N_) \
  switch (rank) { case 4: D6(4, N_); break; case 8: D6(8, N_); break; \
                  case 16: D6(16, N_); break; case 32: D6(32, N_); break; \
                  default: D6(64, N_); break; }
  if (nother == 2) { D6F(2); } else { D6F(3); }
#undef D6F
#undef D6
  const int64_t elems = nrows_out * rank;
  hipLaunchKernelGGL((det6_fold_kern<V>), dim3(512), dim3(256), 0, st,
                     outb, nbuckets, elems, out);
  return 0;
}

inline int64_t det_pick_span(int64_t nnz) {
  const char * e = getenv("SPLATT_SPAN_WAVES");
  const int64_t target_waves = e ? atoll(e) : 65536;
  int64_t span = nnz / target_waves;
  if (span < 256) span = 256;
  if (span > 16384) span = 16384;
  return span;
}

inline int64_t det_nwalkers(int64_t nnz, int rank) {
  const int64_t span = det_pick_span(nnz);
  const int64_t nwaves = (nnz + span - 1) / span;
  return nwaves * (WAVE / rank);
}

inline bool det_spec_ok(int F) {
  return F == 4 || F == 8 || F == 16 || F == 32 || F == 64;
}

template <typename V>
int launch_flat_det(const int32_t * key, const int32_t * const idx[8],
                    const V * const mats[8], const V * vals, int64_t nnz,
                    V * out, V * side, int64_t side_elems, int rank,
                    int nother, hipStream_t st) {
  if (!det_spec_ok(rank) || nother > 4) return -1;
  const int64_t span = det_pick_span(nnz);
  const int64_t nwaves = (nnz + span - 1) / span;
  const int64_t nw = nwaves * (WAVE / rank);
  if (side_elems < nw * 2 * rank) return -2;
  (void)hipMemsetAsync(side, 0, (size_t)(nw * 2 * rank) * sizeof(V), st);

  const int wpb = 4;
  const int64_t nblocks = (nwaves + wpb - 1) / wpb;
  dim3 grid((uint32_t)nblocks), block(wpb * WAVE);

#define DARGS key, idx[0], idx[1], idx[2], idx[3], mats[0], mats[1], \
              mats[2], mats[3], vals, nnz, span, out, side
#define DFIX(F_) \
  { const int tpb = 256; \
    const int64_t fb = (nw + tpb / F_ - 1) / (tpb / F_); \
    hipLaunchKernelGGL((mttkrp_det_fixup_kern<V, F_>), dim3((uint32_t)fb), \
                       dim3(tpb), 0, st, key, nnz, span, nw, side, out); }
#define DK(F_, N_) \
  { hipLaunchKernelGGL((mttkrp_det_kern<V, F_, N_>), grid, block, 0, st, \
                       DARGS); DFIX(F_); }
#define DF(N_) \
  switch (rank) { case 4: DK(4, N_); break; case 8: DK(8, N_); break; \
                  case 16: DK(16, N_); break; case 32: DK(32, N_); break; \
                  default: DK(64, N_); break; }
  switch (nother) {
    case 2: DF(2); break;
    case 3: DF(3); break;
    default: DF(4); break;
  }
#undef DF
#undef DK
#undef DFIX
#undef DARGS
  return 0;
}

}  // namespace

extern "C" int splatt_hip_mttkrp_det6_f64(
    const int32_t * pack, const double * m0, const double * m1,
    const double * m2, const double * vals, const int64_t * blk_start,
    const int64_t * blk_end, const int32_t * blk_row0,
    const int64_t * blk_bucket_p0, int64_t nblocks,
    int32_t chunk, int32_t dim0, int64_t nrows_out, int64_t nbuckets,
    double * outb, double * side, double * out, int rank, int nother,
    void * stream) {
  const double * mats[3] = {m0, m1, m2};
  return launch_det6<double>(pack, mats, vals, blk_start, blk_end,
                             blk_row0, blk_bucket_p0, nblocks, chunk, dim0, nrows_out,
                             nbuckets, outb, side, out, rank, nother,
                             (hipStream_t)stream);
}

extern "C" int splatt_hip_mttkrp_det6_f32(
    const int32_t * pack, const float * m0, const float * m1,
    const float * m2, const float * vals, const int64_t * blk_start,
    const int64_t * blk_end, const int32_t * blk_row0,
    const int64_t * blk_bucket_p0, int64_t nblocks,
    int32_t chunk, int32_t dim0, int64_t nrows_out, int64_t nbuckets,
    float * outb, float * side, float * out, int rank, int nother,
    void * stream) {
  const float * mats[3] = {m0, m1, m2};
  return launch_det6<float>(pack, mats, vals, blk_start, blk_end,
                            blk_row0, blk_bucket_p0, nblocks, chunk, dim0,
                            nrows_out, nbuckets, outb, side, out, rank,
                            nother, (hipStream_t)stream);
}

extern "C" int64_t splatt_hip_flat_det_ws(int64_t nnz, int rank) {
  if (!det_spec_ok(rank)) return -1;
  return det_nwalkers(nnz, rank) * 2 * rank;
}

extern "C" int splatt_hip_mttkrp_flat_det_f64(
    const int32_t * key, const int32_t * const * idx,
    const double * const * mats, const double * vals, int64_t nnz,
    double * out, double * side, int64_t side_elems, int rank, int nother,
    void * stream) {
  return launch_flat_det<double>(key, idx, mats, vals, nnz, out, side,
                                 side_elems, rank, nother,
                                 (hipStream_t)stream);
}

extern "C" int splatt_hip_mttkrp_flat_det_f32(
    const int32_t * key, const int32_t * const * idx,
    const float * const * mats, const float * vals, int64_t nnz,
    float * out, float * side, int64_t side_elems, int rank, int nother,
    void * stream) {
  return launch_flat_det<float>(key, idx, mats, vals, nnz, out, side,
                                side_elems, rank, nother,
                                (hipStream_t)stream);
}
