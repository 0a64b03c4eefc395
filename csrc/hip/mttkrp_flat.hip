// CDNA4 flat ("expanded-CSF") MTTKRP — the default device hot path.
//
// Motivation (measured, profiles/): the hierarchical CSF walk is a serial
// dependent-load chain per fiber; on power-law tensors with short fibers it
// runs latency-bound at ~1.5% of HBM roofline. This kernel linearizes the
// computation: for every nonzero p (in CSF-sorted order)
//     prod[p] = vals[p] * PROD_t mats[t][ idx_t[p] ]     (t = non-out modes)
//     out[key[p], :] += prod[p]
// where idx_t / key are the per-nnz ancestor-label expansions of the CSF
// levels (splatt_amd/csf.py ancestor_expand). Because nonzeros are sorted,
// `key` is piecewise-constant, so each 64/F-lane column group walks a
// contiguous sub-span serially, folds products into a register accumulator,
// and emits one hardware f64/f32 atomic-add per key RUN — not per nonzero.
// The U-deep unrolled loads are independent, so the loop is
// throughput-bound, not latency-bound (the CSF-walk family in
// mttkrp_kernels.hip is kept as a comparison algorithm, reference-style
// `splatt bench -a`).
//
// Capability parity: same operator as reference mttkrp.c:390-1278 at any
// output depth, any nmodes in {3,4,5}, f32/f64, any rank (spec F in
// {4,8,16,32,64}, generic otherwise).
#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstdlib>

#include "store_types.hpp"

namespace {
using splatt_store::bf16;
using splatt_store::to_compute;

constexpr int WAVE = 64;

__device__ __forceinline__ int64_t min64(int64_t a, int64_t b) { return a < b ? a : b; }

template <typename V>
__device__ __forceinline__ void atomic_add_g(V * p, V v) {
  unsafeAtomicAdd(p, v);
}

// streamed-once arrays (key/idx/vals) are loaded non-temporally so they do
// not evict the factor-row working set from the per-XCD L2s
template <typename T>
__device__ __forceinline__ T ldnt(const T * p) {
  return __builtin_nontemporal_load(p);
}

// ---------------------------------------------------------- spec kernels
// F lanes per column group, R = 64/F groups each walking a contiguous
// sub-span; U-deep unroll for memory-level parallelism.
template <typename V, int F, int NOTHER, int U = 8>
__global__ void __launch_bounds__(256)
mttkrp_flat_kern(const int32_t * __restrict__ key,
                 const int32_t * __restrict__ i0,
                 const int32_t * __restrict__ i1,
                 const int32_t * __restrict__ i2,
                 const int32_t * __restrict__ i3,
                 const V * __restrict__ m0, const V * __restrict__ m1,
                 const V * __restrict__ m2, const V * __restrict__ m3,
                 const V * __restrict__ vals, int64_t nnz, int64_t span,
                 V * __restrict__ out) {
  constexpr int R = WAVE / F;
  const int lane = threadIdx.x & (WAVE - 1);
  const int64_t wid = (int64_t)blockIdx.x * (blockDim.x / WAVE)
                      + (threadIdx.x / WAVE);
  const int c = lane % F;
  const int g = lane / F;
  const int64_t w0 = wid * span;
  if (w0 >= nnz) return;
  const int64_t w1 = min64(nnz, w0 + span);
  // contiguous sub-span per column group
  const int64_t gsz = (w1 - w0 + R - 1) / R;
  const int64_t p0 = min64(w1, w0 + g * gsz);
  const int64_t p1 = min64(w1, p0 + gsz);
  if (p0 >= p1) return;

  int32_t cur = key[p0];
  V acc = (V)0;
  int64_t p = p0;
  while (p < p1) {
    const int n = (int)min64((int64_t)U, p1 - p);
    int32_t k[U];
    V prod[U];
    if (n == U) {
      #pragma unroll
      for (int u = 0; u < U; ++u) k[u] = ldnt(&key[p + u]);
      #pragma unroll
      for (int u = 0; u < U; ++u) {
        V x = ldnt(&vals[p + u]) * m0[(int64_t)ldnt(&i0[p + u]) * F + c]
                          * m1[(int64_t)ldnt(&i1[p + u]) * F + c];
        if (NOTHER > 2) x *= m2[(int64_t)ldnt(&i2[p + u]) * F + c];
        if (NOTHER > 3) x *= m3[(int64_t)ldnt(&i3[p + u]) * F + c];
        prod[u] = x;
      }
    } else {
      #pragma unroll
      for (int u = 0; u < U; ++u) {
        if (u < n) {
          k[u] = ldnt(&key[p + u]);
          V x = ldnt(&vals[p + u]) * m0[(int64_t)ldnt(&i0[p + u]) * F + c]
                            * m1[(int64_t)ldnt(&i1[p + u]) * F + c];
          if (NOTHER > 2) x *= m2[(int64_t)ldnt(&i2[p + u]) * F + c];
          if (NOTHER > 3) x *= m3[(int64_t)ldnt(&i3[p + u]) * F + c];
          prod[u] = x;
        } else {
          k[u] = cur;        // no-op in the fold
          prod[u] = (V)0;
        }
      }
    }
    #pragma unroll
    for (int u = 0; u < U; ++u) {
      if (k[u] != cur) {     // rare for root output; per-run for intl/leaf
        atomic_add_g(&out[(int64_t)cur * F + c], acc);
        acc = (V)0;
        cur = k[u];
      }
      acc += prod[u];
    }
    p += n;
  }
  atomic_add_g(&out[(int64_t)cur * F + c], acc);
}

// ------------------------------------------------- staged spec kernel (v2)
// v1 has every lane of a column group redundantly load the same stream
// words (key/idx/vals), so memory-level parallelism is capped by the VGPR
// cost of the unroll. v2 stages streams COALESCED — lane slot c of a group
// loads element pb+c of each stream in one instruction per F nonzeros —
// and redistributes them with ds_bpermute (__shfl); the factor-row gathers
// then issue in independent batches of 8, giving ~4x the outstanding
// gathers per wave at lower VGPR pressure.
template <typename V, int F, int NOTHER, int GBP = 8, typename S = V>
__global__ void __launch_bounds__(256)
mttkrp_flat2_kern(const int32_t * __restrict__ key,
                  const int32_t * __restrict__ i0,
                  const int32_t * __restrict__ i1,
                  const int32_t * __restrict__ i2,
                  const int32_t * __restrict__ i3,
                  const S * __restrict__ m0, const S * __restrict__ m1,
                  const S * __restrict__ m2, const S * __restrict__ m3,
                  const V * __restrict__ vals, int64_t nnz, int64_t span,
                  V * __restrict__ out) {
  constexpr int R = WAVE / F;
  constexpr int GB = (F >= GBP) ? GBP : F;   // gather sub-batch
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

  int32_t cur = key[p0];
  V acc = (V)0;
  for (int64_t pb = p0; pb < p1; pb += F) {
    const int nb = (int)min64((int64_t)F, p1 - pb);       // group-uniform
    const int64_t ps = pb + (c < nb ? c : nb - 1);        // clamped slot
    const int32_t kreg = ldnt(&key[ps]);
    const int32_t i0reg = ldnt(&i0[ps]);
    const int32_t i1reg = ldnt(&i1[ps]);
    const int32_t i2reg = (NOTHER > 2) ? ldnt(&i2[ps]) : 0;
    const int32_t i3reg = (NOTHER > 3) ? ldnt(&i3[ps]) : 0;
    const V vreg = ldnt(&vals[ps]);
    for (int ub = 0; ub < nb; ub += GB) {
      const int ne = nb - ub < GB ? nb - ub : GB;         // group-uniform
      int32_t kk[GB];
      V vv[GB];
      S a0[GB], a1[GB], a2[GB], a3[GB];
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
        V x = vv[u] * to_compute(a0[u], (V)0) * to_compute(a1[u], (V)0);
        if (NOTHER > 2) x *= to_compute(a2[u], (V)0);
        if (NOTHER > 3) x *= to_compute(a3[u], (V)0);
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

// storage-typed launcher for the v2 kernel (the default spec path):
// reduced-precision factor STORE for non-staged HBM-bound dispatches
inline int64_t pick_span_decl(int64_t nnz);
template <typename V, typename S>
void launch_flat_store(const int32_t * key, const int32_t * const idx[8],
                       const S * const mats[8], const V * vals, int64_t nnz,
                       V * out, int rank, int nother, hipStream_t st) {
  const int64_t span = pick_span_decl(nnz);
  const int64_t nwaves = (nnz + span - 1) / span;
  const int wpb = 4;
  const int64_t nblocks = (nwaves + wpb - 1) / wpb;
  dim3 grid((uint32_t)nblocks), block(wpb * WAVE);
#define SARGS key, idx[0], idx[1], idx[2], idx[3], mats[0], mats[1], \
              mats[2], mats[3], vals, nnz, span, out
#define LS(F_, N_) \
  hipLaunchKernelGGL((mttkrp_flat2_kern<V, F_, N_, 8, S>), grid, block, 0, \
                     st, SARGS)
#define LSF(N_) \
  switch (rank) { case 4: LS(4, N_); break; case 8: LS(8, N_); break; \
                  case 16: LS(16, N_); break; case 32: LS(32, N_); break; \
                  default: LS(64, N_); break; }
  switch (nother) {
    case 2: LSF(2); break;
    case 3: LSF(3); break;
    default: LSF(4); break;
  }
#undef LSF
#undef LS
#undef SARGS
}

// ------------------------------------------- pipelined spec kernel (v3)
// v2 with double-buffered gather batches: batch k+1's factor-row gathers
// are ISSUED before batch k is folded, so the fold overlaps the next
// batch's memory latency (hipcc emits counted s_waitcnt instead of a full
// drain per batch). Costs ~2x the batch registers.
template <typename V, int F, int NOTHER>
__global__ void __launch_bounds__(256)
mttkrp_flat3_kern(const int32_t * __restrict__ key,
                  const int32_t * __restrict__ i0,
                  const int32_t * __restrict__ i1,
                  const int32_t * __restrict__ i2,
                  const int32_t * __restrict__ i3,
                  const V * __restrict__ m0, const V * __restrict__ m1,
                  const V * __restrict__ m2, const V * __restrict__ m3,
                  const V * __restrict__ vals, int64_t nnz, int64_t span,
                  V * __restrict__ out) {
  constexpr int R = WAVE / F;
  constexpr int GB = (F >= 8) ? 8 : F;
  constexpr int NB = (F + GB - 1) / GB;   // batches per stream window
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

  int32_t cur = key[p0];
  V acc = (V)0;
  int32_t kk[2][GB];
  V vv[2][GB], a0[2][GB], a1[2][GB], a2[2][GB], a3[2][GB];

  for (int64_t pb = p0; pb < p1; pb += F) {
    const int nb = (int)min64((int64_t)F, p1 - pb);
    const int64_t ps = pb + (c < nb ? c : nb - 1);
    const int32_t kreg = ldnt(&key[ps]);
    const int32_t i0reg = ldnt(&i0[ps]);
    const int32_t i1reg = ldnt(&i1[ps]);
    const int32_t i2reg = (NOTHER > 2) ? ldnt(&i2[ps]) : 0;
    const int32_t i3reg = (NOTHER > 3) ? ldnt(&i3[ps]) : 0;
    const V vreg = ldnt(&vals[ps]);

    auto issue = [&](int buf, int ub, int ne) {
      #pragma unroll
      for (int u = 0; u < GB; ++u) {
        const int src = gbase + (u < ne ? ub + u : ub);
        kk[buf][u] = __shfl(kreg, src, WAVE);
        vv[buf][u] = __shfl(vreg, src, WAVE);
        const int32_t j0 = __shfl(i0reg, src, WAVE);
        const int32_t j1 = __shfl(i1reg, src, WAVE);
        a0[buf][u] = m0[(int64_t)j0 * F + c];
        a1[buf][u] = m1[(int64_t)j1 * F + c];
        if (NOTHER > 2) {
          const int32_t j2 = __shfl(i2reg, src, WAVE);
          a2[buf][u] = m2[(int64_t)j2 * F + c];
        }
        if (NOTHER > 3) {
          const int32_t j3 = __shfl(i3reg, src, WAVE);
          a3[buf][u] = m3[(int64_t)j3 * F + c];
        }
      }
    };
    auto fold = [&](int buf, int ne) {
      #pragma unroll
      for (int u = 0; u < GB; ++u) {
        if (u >= ne) break;
        V x = vv[buf][u] * a0[buf][u] * a1[buf][u];
        if (NOTHER > 2) x *= a2[buf][u];
        if (NOTHER > 3) x *= a3[buf][u];
        if (kk[buf][u] != cur) {
          atomic_add_g(&out[(int64_t)cur * F + c], acc);
          acc = (V)0;
          cur = kk[buf][u];
        }
        acc += x;
      }
    };

    int ne_of[NB];
    int nbat = 0;
    for (int ub = 0; ub < nb; ub += GB)
      ne_of[nbat++] = nb - ub < GB ? nb - ub : GB;
    // issue batch b+1 before folding batch b
    issue(0, 0, ne_of[0]);
    for (int b = 0; b < nbat; ++b) {
      if (b + 1 < nbat) issue((b + 1) & 1, (b + 1) * GB, ne_of[b + 1]);
      fold(b & 1, ne_of[b]);
    }
  }
  atomic_add_g(&out[(int64_t)cur * F + c], acc);
}

// ----------------------------------------- vectorized-gather kernel (v4)
// PMC evidence (profiles/): v2 runs at TA_BUSY ~87% — the per-CU vector
// memory ADDRESS path is the bottleneck, not bandwidth or occupancy. v4
// halves the address count: each lane gathers 16 B (VW = 16/sizeof(V)
// columns) per factor row, so a row of F columns needs F/VW lane
// addresses instead of F. Lane layout: L = F/VW lanes per column group,
// R = 64/L groups; each lane folds VW columns in registers.
template <typename V, int F, int NOTHER>
__global__ void __launch_bounds__(256)
mttkrp_flat4_kern(const int32_t * __restrict__ key,
                  const int32_t * __restrict__ i0,
                  const int32_t * __restrict__ i1,
                  const int32_t * __restrict__ i2,
                  const int32_t * __restrict__ i3,
                  const V * __restrict__ m0, const V * __restrict__ m1,
                  const V * __restrict__ m2, const V * __restrict__ m3,
                  const V * __restrict__ vals, int64_t nnz, int64_t span,
                  V * __restrict__ out) {
  constexpr int VW = 16 / sizeof(V);        // columns per lane (16B loads)
  constexpr int L = (F + VW - 1) / VW;      // lanes per column group
  constexpr int R = WAVE / L;               // groups per wave
  constexpr int GB = (NOTHER > 2) ? 4 : 6;  // gather batch (register budget)
  using Vec = __attribute__((ext_vector_type(VW))) V;
  const int lane = threadIdx.x & (WAVE - 1);
  const int64_t wid = (int64_t)blockIdx.x * (blockDim.x / WAVE)
                      + (threadIdx.x / WAVE);
  const int li = lane % L;                  // lane within group
  const int c0 = li * VW;                   // first column of this lane
  const int g = lane / L;
  const int gbase = g * L;
  const int64_t w0 = wid * span;
  if (w0 >= nnz) return;
  const int64_t w1 = min64(nnz, w0 + span);
  const int64_t gsz = (w1 - w0 + R - 1) / R;
  const int64_t p0 = min64(w1, w0 + g * gsz);
  const int64_t p1 = min64(w1, p0 + gsz);
  if (p0 >= p1) return;

  int32_t cur = key[p0];
  V acc[VW];
  #pragma unroll
  for (int w = 0; w < VW; ++w) acc[w] = (V)0;

  for (int64_t pb = p0; pb < p1; pb += L) {
    const int nb = (int)min64((int64_t)L, p1 - pb);   // stream window = L
    const int64_t ps = pb + (li < nb ? li : nb - 1);
    const int32_t kreg = ldnt(&key[ps]);
    const int32_t i0reg = ldnt(&i0[ps]);
    const int32_t i1reg = ldnt(&i1[ps]);
    const int32_t i2reg = (NOTHER > 2) ? ldnt(&i2[ps]) : 0;
    const int32_t i3reg = (NOTHER > 3) ? ldnt(&i3[ps]) : 0;
    const V vreg = ldnt(&vals[ps]);
    for (int ub = 0; ub < nb; ub += GB) {
      const int ne = nb - ub < GB ? nb - ub : GB;
      int32_t kk[GB];
      V vv[GB];
      Vec a0[GB], a1[GB], a2[GB], a3[GB];
      #pragma unroll
      for (int u = 0; u < GB; ++u) {
        const int src = gbase + (u < ne ? ub + u : ub);
        kk[u] = __shfl(kreg, src, WAVE);
        vv[u] = __shfl(vreg, src, WAVE);
        const int32_t j0 = __shfl(i0reg, src, WAVE);
        const int32_t j1 = __shfl(i1reg, src, WAVE);
        a0[u] = *reinterpret_cast<const Vec*>(&m0[(int64_t)j0 * F + c0]);
        a1[u] = *reinterpret_cast<const Vec*>(&m1[(int64_t)j1 * F + c0]);
        if (NOTHER > 2) {
          const int32_t j2 = __shfl(i2reg, src, WAVE);
          a2[u] = *reinterpret_cast<const Vec*>(&m2[(int64_t)j2 * F + c0]);
        }
        if (NOTHER > 3) {
          const int32_t j3 = __shfl(i3reg, src, WAVE);
          a3[u] = *reinterpret_cast<const Vec*>(&m3[(int64_t)j3 * F + c0]);
        }
      }
      #pragma unroll
      for (int u = 0; u < GB; ++u) {
        if (u >= ne) break;
        if (kk[u] != cur) {
          #pragma unroll
          for (int w = 0; w < VW; ++w) {
            atomic_add_g(&out[(int64_t)cur * F + c0 + w], acc[w]);
            acc[w] = (V)0;
          }
          cur = kk[u];
        }
        #pragma unroll
        for (int w = 0; w < VW; ++w) {
          V x = vv[u] * a0[u][w] * a1[u][w];
          if (NOTHER > 2) x *= a2[u][w];
          if (NOTHER > 3) x *= a3[u][w];
          acc[w] += x;
        }
      }
    }
  }
  #pragma unroll
  for (int w = 0; w < VW; ++w)
    atomic_add_g(&out[(int64_t)cur * F + c0 + w], acc[w]);
}

// ------------------------------------------------------ generic-rank kernel
// lane = column (chunked by 64), wave walks its span serially. Correctness
// path for ranks outside the spec set.
// up to 7 other modes (8-mode tensors, reference SPLATT_MAX_NMODES);
// idx/mats arrive BY VALUE in the kernel-arg struct (no device-side table,
// no upload: the launch stays async and hipGraph-capturable)
template <typename V>
struct PtrTab {
  const int32_t * idx[8];
  const V * mats[8];
};

template <typename V, int NOTHER>
__global__ void __launch_bounds__(256)
mttkrp_flat_generic_kern(const int32_t * __restrict__ key,
                         const PtrTab<V> tab,
                         const V * __restrict__ vals, int64_t nnz,
                         int64_t span, int rank, V * __restrict__ out) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int64_t wid = (int64_t)blockIdx.x * (blockDim.x / WAVE)
                      + (threadIdx.x / WAVE);
  const int64_t p0 = wid * span;
  if (p0 >= nnz) return;
  const int64_t p1 = min64(nnz, p0 + span);
  const int32_t * ip[NOTHER];
  const V * mp[NOTHER];
  #pragma unroll
  for (int t = 0; t < NOTHER; ++t) { ip[t] = tab.idx[t]; mp[t] = tab.mats[t]; }
  for (int cb = 0; cb < rank; cb += WAVE) {
    const int c = cb + lane;
    if (c >= rank) break;
    int32_t cur = key[p0];
    V acc = (V)0;
    for (int64_t p = p0; p < p1; ++p) {
      V x = vals[p];
      #pragma unroll
      for (int t = 0; t < NOTHER; ++t)
        x *= mp[t][(int64_t)ip[t][p] * rank + c];
      const int32_t k = key[p];
      if (k != cur) {
        atomic_add_g(&out[(int64_t)cur * rank + c], acc);
        acc = (V)0;
        cur = k;
      }
      acc += x;
    }
    atomic_add_g(&out[(int64_t)cur * rank + c], acc);
  }
}

inline int64_t pick_span(int64_t nnz) {
  const char * e = getenv("SPLATT_SPAN_WAVES");   // tuning knob
  const int64_t target_waves = e ? atoll(e) : 65536;
  int64_t span = nnz / target_waves;
  if (span < 256) span = 256;
  if (span > 16384) span = 16384;
  return span;
}

inline int64_t pick_span_decl(int64_t nnz) { return pick_span(nnz); }

inline bool spec_ok(int F) {
  return F == 4 || F == 8 || F == 16 || F == 32 || F == 64;
}

inline int pick_unroll() {
  // A/B lever: 0 (default) = staged v2 kernel; 4/8/16 = v1 at that unroll
  const char * e = getenv("SPLATT_MTTKRP_U");
  const int u = e ? atoi(e) : 0;
  return (u == 2 || u == 3 || u == 4 || u == 5 || u == 8 || u == 16) ? u : 0;
}

template <typename V>
void launch_flat(const int32_t * key, const int32_t * const idx[8],
                 const V * const mats[8], const V * vals, int64_t nnz,
                 V * out, int rank, int nother, hipStream_t st) {
  const int64_t span = pick_span(nnz);
  const int64_t nwaves = (nnz + span - 1) / span;
  const int wpb = 4;
  const int64_t nblocks = (nwaves + wpb - 1) / wpb;
  dim3 grid((uint32_t)nblocks), block(wpb * WAVE);

#define ARGS key, idx[0], idx[1], idx[2], idx[3], mats[0], mats[1], mats[2], \
             mats[3], vals, nnz, span, out
#define L1(F_, N_, U_) \
  hipLaunchKernelGGL((mttkrp_flat_kern<V, F_, N_, U_>), grid, block, 0, st, ARGS)
#define L2K(F_, N_) \
  hipLaunchKernelGGL((mttkrp_flat2_kern<V, F_, N_>), grid, block, 0, st, ARGS)
#define L2K4(F_, N_) \
  hipLaunchKernelGGL((mttkrp_flat2_kern<V, F_, N_, 4>), grid, block, 0, st, ARGS)
#define L3K(F_, N_) \
  hipLaunchKernelGGL((mttkrp_flat3_kern<V, F_, N_>), grid, block, 0, st, ARGS)
#define L4K(F_, N_) \
  hipLaunchKernelGGL((mttkrp_flat4_kern<V, F_, N_>), grid, block, 0, st, ARGS)
#define LU(F_, N_) \
  switch (uu) { case 4: L1(F_, N_, 4); break; case 16: L1(F_, N_, 16); break; \
                case 8: L1(F_, N_, 8); break; case 3: L3K(F_, N_); break; \
                case 2: L2K4(F_, N_); break; case 5: L4K(F_, N_); break; \
                default: L2K(F_, N_); break; }
#define LF(N_) \
  switch (rank) { case 4: LU(4, N_); break; case 8: LU(8, N_); break; \
                  case 16: LU(16, N_); break; case 32: LU(32, N_); break; \
                  default: LU(64, N_); break; }
  if (spec_ok(rank) && nother <= 4) {
    const int uu = pick_unroll();
    switch (nother) {
      case 2: LF(2); break;
      case 3: LF(3); break;
      default: LF(4); break;
    }
    return;
  }
#undef LF
#undef LU
#undef L1
#undef ARGS
  // generic path (odd ranks or >5 modes): pointer tables travel by value
  // in the kernel-arg block — async launch, graph-capturable, no cleanup
  PtrTab<V> tab{};
  for (int t = 0; t < nother; ++t) { tab.idx[t] = idx[t]; tab.mats[t] = mats[t]; }
#define GARGS key, tab, vals, nnz, span, rank, out
  switch (nother) {
    case 2:  hipLaunchKernelGGL((mttkrp_flat_generic_kern<V, 2>), grid, block, 0, st, GARGS); break;
    case 3:  hipLaunchKernelGGL((mttkrp_flat_generic_kern<V, 3>), grid, block, 0, st, GARGS); break;
    case 4:  hipLaunchKernelGGL((mttkrp_flat_generic_kern<V, 4>), grid, block, 0, st, GARGS); break;
    case 5:  hipLaunchKernelGGL((mttkrp_flat_generic_kern<V, 5>), grid, block, 0, st, GARGS); break;
    case 6:  hipLaunchKernelGGL((mttkrp_flat_generic_kern<V, 6>), grid, block, 0, st, GARGS); break;
    default: hipLaunchKernelGGL((mttkrp_flat_generic_kern<V, 7>), grid, block, 0, st, GARGS); break;
  }
#undef GARGS
}

}  // namespace

extern "C" void splatt_hip_mttkrp_flat_f64(
    const int32_t * key, const int32_t * const * idx,
    const double * const * mats,
    const double * vals, int64_t nnz, double * out, int rank, int nother,
    void * stream) {
  launch_flat<double>(key, idx, mats, vals, nnz, out, rank, nother,
                      (hipStream_t)stream);
}

extern "C" void splatt_hip_mttkrp_flat_f32(
    const int32_t * key, const int32_t * const * idx,
    const float * const * mats,
    const float * vals, int64_t nnz, float * out, int rank, int nother,
    void * stream) {
  launch_flat<float>(key, idx, mats, vals, nnz, out, rank, nother,
                     (hipStream_t)stream);
}

// reduced-precision factor STORAGE (f64 accumulate) on the v2 path;
// returns nonzero for unsupported (non-spec) ranks
extern "C" int splatt_hip_mttkrp_flat_f64f32(
    const int32_t * key, const int32_t * const * idx,
    const float * const * mats, const double * vals, int64_t nnz,
    double * out, int rank, int nother, void * stream) {
  if (!spec_ok(rank) || nother > 4) return -1;
  launch_flat_store<double, float>(key, idx, mats, vals, nnz, out, rank,
                                   nother, (hipStream_t)stream);
  return 0;
}

extern "C" int splatt_hip_mttkrp_flat_f64bf16(
    const int32_t * key, const int32_t * const * idx,
    const uint16_t * const * mats, const double * vals, int64_t nnz,
    double * out, int rank, int nother, void * stream) {
  if (!spec_ok(rank) || nother > 4) return -1;
  launch_flat_store<double, bf16>(key, idx,
                                  (const bf16 * const *)mats, vals, nnz,
                                  out, rank, nother, (hipStream_t)stream);
  return 0;
}
