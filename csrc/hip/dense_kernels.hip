// Small dense kernels for the ALS normal-equations side on gfx950.
//
// gram: G = A^T A for tall-skinny row-major A (n x F, F <= 64).
// rocBLAS/Tensile picks a one-workgroup macro-tile for M=N=F<=64 with huge
// K (measured 1.5 ms for 29k x 16 — profiles/), so we do the reduction
// ourselves: each block stages TR=32 rows through LDS, every thread owns
// ceil(F^2/256) output pairs, partials land in G with one atomic per pair
// per block. Bandwidth-bound by design: A is read exactly once.
// Capability parity: reference mat_aTa (src/matrix.c:414-455).
#include <hip/hip_runtime.h>
#include <cstdint>

namespace {

constexpr int TPB = 256;
constexpr int TR = 32;  // rows staged per LDS tile

template <typename V>
__device__ __forceinline__ void atomic_add_g(V * p, V v) {
  unsafeAtomicAdd(p, v);
}

template <typename V, int PAIRS>
__global__ void __launch_bounds__(TPB)
gram_kern(const V * __restrict__ A, int64_t n, int F, int64_t rows_per_blk,
          V * __restrict__ G) {
  __shared__ V tile[TR * 64];
  const int tid = threadIdx.x;
  const int FF = F * F;
  const int64_t r0 = (int64_t)blockIdx.x * rows_per_blk;
  if (r0 >= n) return;
  const int64_t r1 = min(n, r0 + rows_per_blk);

  V acc[PAIRS];
  #pragma unroll
  for (int q = 0; q < PAIRS; ++q) acc[q] = (V)0;

  for (int64_t rt = r0; rt < r1; rt += TR) {
    const int nr = (int)min((int64_t)TR, r1 - rt);
    __syncthreads();
    for (int e = tid; e < nr * F; e += TPB) tile[e] = A[rt * F + e];
    __syncthreads();
    for (int r = 0; r < nr; ++r) {
      const V * row = tile + r * F;
      #pragma unroll
      for (int q = 0; q < PAIRS; ++q) {
        const int idx = tid + q * TPB;
        if (idx < FF) acc[q] += row[idx / F] * row[idx % F];
      }
    }
  }
  #pragma unroll
  for (int q = 0; q < PAIRS; ++q) {
    const int idx = tid + q * TPB;
    if (idx < FF) atomic_add_g(&G[idx], acc[q]);
  }
}

// ------------------------------------------------------------ MFMA gram
// G = A^T A on the matrix cores (v_mfma_f64_16x16x4_f64 /
// v_mfma_f32_16x16x4_f32). Key trick: for these shapes the A-operand
// layout is A[m=lane&15][k=lane>>4] and the B-operand layout is
// B[k=lane>>4][n=lane&15] — mutual transposes — so feeding the SAME
// per-lane value x = chunk[k][col] as both operands computes
// chunk^T * chunk directly. Each wave folds its row range 4 rows per
// MFMA into (F/16)^2 16x16 accumulator tiles (upper-triangular block
// pairs only; the mirror is written in the epilogue).
template <typename V> struct MfmaAcc;
template <> struct MfmaAcc<double> {
  using type = __attribute__((ext_vector_type(4))) double;
  static __device__ __forceinline__ type mfma(double a, double b, type c) {
    return __builtin_amdgcn_mfma_f64_16x16x4f64(a, b, c, 0, 0, 0);
  }
  // f64 16x16x4 C/D row map (probed on gfx950): row = lane/16 + 4*reg
  static __device__ __forceinline__ int crow(int lane, int i) {
    return (lane >> 4) + 4 * i;
  }
};
template <> struct MfmaAcc<float> {
  using type = __attribute__((ext_vector_type(4))) float;
  static __device__ __forceinline__ type mfma(float a, float b, type c) {
    return __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, c, 0, 0, 0);
  }
  // f32 16x16x4 uses the standard map: row = (lane/16)*4 + reg
  static __device__ __forceinline__ int crow(int lane, int i) {
    return (lane >> 4) * 4 + i;
  }
};

template <typename V, int F>
__global__ void __launch_bounds__(256)
gram_mfma_kern(const V * __restrict__ A, int64_t n, int64_t rows_per_blk,
               V * __restrict__ G) {
  constexpr int NB = F / 16;              // 16-wide column blocks
  constexpr int NP = NB * (NB + 1) / 2;   // upper-triangular block pairs
  using Acc = typename MfmaAcc<V>::type;
  const int lane = threadIdx.x & 63;
  const int wv = threadIdx.x >> 6;
  const int c = lane & 15;                // column within a 16-block
  const int k = lane >> 4;                // row within the K=4 chunk
  const int64_t r0 = (int64_t)blockIdx.x * rows_per_blk + (int64_t)wv * 4;
  const int64_t r1 = min((int64_t)blockIdx.x * rows_per_blk + rows_per_blk, n);

  Acc acc[NP];
  #pragma unroll
  for (int p = 0; p < NP; ++p) acc[p] = Acc{};

  for (int64_t r = r0; r < r1; r += 4 * 4 /* waves */) {
    V v[NB];
    const int64_t row = r + k;
    #pragma unroll
    for (int b = 0; b < NB; ++b)
      v[b] = row < r1 ? A[row * F + b * 16 + c] : (V)0;
    int p = 0;
    #pragma unroll
    for (int bi = 0; bi < NB; ++bi) {
      #pragma unroll
      for (int bj = bi; bj < NB; ++bj) {
        acc[p] = MfmaAcc<V>::mfma(v[bi], v[bj], acc[p]);
        ++p;
      }
    }
  }

  // C/D layout of the 16x16x4 forms (probed on gfx950, profiles/):
  // reg i -> row (lane>>4) + 4*i, col lane&15
  int p = 0;
  #pragma unroll
  for (int bi = 0; bi < NB; ++bi) {
    #pragma unroll
    for (int bj = bi; bj < NB; ++bj, ++p) {
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int row = bi * 16 + MfmaAcc<V>::crow(lane, i);
        const int col = bj * 16 + c;
        atomic_add_g(&G[row * F + col], acc[p][i]);
        if (bi != bj) atomic_add_g(&G[col * F + row], acc[p][i]);
      }
    }
  }
}

template <typename V>
void launch_gram(const V * A, int64_t n, int F, V * G, hipStream_t st) {
  // enough blocks to fill the chip, >= 4 LDS tiles of work each
  int64_t nblocks = (n + TR * 4 - 1) / (TR * 4);
  if (nblocks > 2048) nblocks = 2048;
  if (nblocks < 1) nblocks = 1;
  const int64_t rows_per_blk = (n + nblocks - 1) / nblocks;
  dim3 grid((uint32_t)nblocks), block(TPB);
  // matrix-core path for the MFMA-tileable ranks
  if (F == 16 || F == 32 || F == 64) {
    switch (F) {
      case 16: hipLaunchKernelGGL((gram_mfma_kern<V, 16>), grid, block, 0, st, A, n, rows_per_blk, G); return;
      case 32: hipLaunchKernelGGL((gram_mfma_kern<V, 32>), grid, block, 0, st, A, n, rows_per_blk, G); return;
      default: hipLaunchKernelGGL((gram_mfma_kern<V, 64>), grid, block, 0, st, A, n, rows_per_blk, G); return;
    }
  }
  const int ff = F * F;
  if (ff <= 256)
    hipLaunchKernelGGL((gram_kern<V, 1>), grid, block, 0, st, A, n, F, rows_per_blk, G);
  else if (ff <= 1024)
    hipLaunchKernelGGL((gram_kern<V, 4>), grid, block, 0, st, A, n, F, rows_per_blk, G);
  else
    hipLaunchKernelGGL((gram_kern<V, 16>), grid, block, 0, st, A, n, F, rows_per_blk, G);
}

// spd_inverse: Ginv = G^-1 for SPD G (F x F, F <= 64) in ONE workgroup —
// Cholesky G = L L^T, triangular inverse X = L^-1, Ginv = X^T X, all in
// LDS. Replaces the rocSOLVER potrf/potri chain (~15 launches + a host
// info sync that breaks hipGraph capture) for the ALS normal equations.
// On breakdown (non-positive pivot) retries with escalating Tikhonov
// jitter, like the host solver (csrc/core/matrix.cpp solve_normals).
template <typename V>
__global__ void __launch_bounds__(64)
spd_inverse_kern(const V * __restrict__ G, V * __restrict__ Ginv, int F) {
  // one 32KB LDS array; Ginv doubles as scratch for X = L^-1
  __shared__ V Lm[64 * 64];
  __shared__ V diagmax_sh;
  __shared__ int fail_sh;
  const int tid = threadIdx.x;

  V jitter = (V)0;
  if (tid == 0) {
    V mx = (V)0;
    for (int j = 0; j < F; ++j) {
      const V d = G[j * F + j];
      mx = mx > d ? mx : d;
    }
    diagmax_sh = mx;
  }
  __syncthreads();
  const V diagmax = diagmax_sh;

  for (int attempt = 0; attempt < 24; ++attempt) {
    for (int e = tid; e < F * F; e += 64) Lm[e] = G[e];
    if (tid == 0) fail_sh = 0;
    __syncthreads();
    if (jitter > (V)0 && tid < F) Lm[tid * F + tid] += jitter;
    __syncthreads();
    // Cholesky, column by column; lanes parallel over rows
    for (int j = 0; j < F; ++j) {
      if (tid == 0) {
        V d = Lm[j * F + j];
        for (int k = 0; k < j; ++k) d -= Lm[j * F + k] * Lm[j * F + k];
        if (d <= (V)0) { fail_sh = 1; d = (V)1; }
        Lm[j * F + j] = sqrt(d);
      }
      __syncthreads();
      if (fail_sh) break;
      const V dj = Lm[j * F + j];
      for (int i = j + 1 + tid; i < F; i += 64) {
        V s = Lm[i * F + j];
        for (int k = 0; k < j; ++k) s -= Lm[i * F + k] * Lm[j * F + k];
        Lm[i * F + j] = s / dj;
      }
      __syncthreads();
    }
    if (!fail_sh) break;
    jitter = (jitter == (V)0) ? diagmax * (V)1e-12 : jitter * (V)100;
    __syncthreads();
  }

  // X = L^-1 (lower): forward substitution, lanes parallel over columns.
  // X lands in the GLOBAL output buffer (scratch), with per-lane private
  // column traffic only — no cross-lane reads of X.
  for (int c = tid; c < F; c += 64) {
    V xcol[64];
    for (int i = c; i < F; ++i) {
      if (i == c) {
        xcol[i] = (V)1 / Lm[i * F + i];
      } else {
        V s = (V)0;
        for (int k = c; k < i; ++k) s += Lm[i * F + k] * xcol[k];
        xcol[i] = -s / Lm[i * F + i];
      }
      Ginv[i * F + c] = xcol[i];
    }
    for (int i = 0; i < c; ++i) Ginv[i * F + c] = (V)0;
  }
  __syncthreads();
  // pull X into LDS (L no longer needed), then Ginv = X^T X
  for (int e = tid; e < F * F; e += 64) Lm[e] = Ginv[e];
  __syncthreads();
  for (int e = tid; e < F * F; e += 64) {
    const int i = e / F, j = e % F;
    const int k0 = i > j ? i : j;
    V s = (V)0;
    for (int k = k0; k < F; ++k) s += Lm[k * F + i] * Lm[k * F + j];
    Ginv[e] = s;
  }
}

// C[n,F] = A[n,F] @ B[F,F] with B staged in LDS — the CPD solve GEMM.
// One thread per output element, a fixed F-length dot: no atomics, no
// split-K, bitwise-deterministic (used when SPLATT_DETERMINISTIC=1; the
// default path keeps the library GEMM). A[r,k] is broadcast across the F
// column lanes of a row, so traffic is ~one read of A + one write of C.
template <typename V, int F>
__global__ void __launch_bounds__(256)
rowsolve_kern(const V * __restrict__ A, const V * __restrict__ B,
              V * __restrict__ C, int64_t n) {
  __shared__ V Bs[F * F];
  for (int i = threadIdx.x; i < F * F; i += blockDim.x) Bs[i] = B[i];
  __syncthreads();
  const int c = threadIdx.x % F;
  const int64_t rstride = (int64_t)gridDim.x * (blockDim.x / F);
  int64_t r = (int64_t)blockIdx.x * (blockDim.x / F) + threadIdx.x / F;
  for (; r < n; r += rstride) {
    V acc = (V)0;
    #pragma unroll
    for (int k = 0; k < F; ++k) acc += A[r * F + k] * Bs[k * F + c];
    C[r * F + c] = acc;
  }
}

// Deterministic G = A^T A, two stages, no atomics: stage 1 gives each
// block a fixed row range and a PRIVATE F x F partial (plain stores);
// stage 2 folds the partials in block order with one workgroup. The
// reduction tree is a pure function of (n, F), so results are bitwise
// run-to-run identical. (A one-workgroup serial version measured 5.3 ms
// at n=29k — a latency-bound dependent chain; splitting across blocks
// recovers the parallelism without reintroducing atomics.)
template <typename V>
__global__ void __launch_bounds__(256)
gram_det_part_kern(const V * __restrict__ A, int64_t n, int F,
                   int64_t rows_per_blk, V * __restrict__ Gpart) {
  const int e = threadIdx.x;
  if (e >= F * F) return;
  const int i = e / F, j = e % F;
  const int64_t r0 = (int64_t)blockIdx.x * rows_per_blk;
  const int64_t r1 = r0 + rows_per_blk < n ? r0 + rows_per_blk : n;
  V acc = (V)0;
  int64_t r = r0;
  for (; r + 4 <= r1; r += 4) {
    // four independent products -> the loads overlap the FMA chain
    const V a0 = A[r * F + i] * A[r * F + j];
    const V a1 = A[(r + 1) * F + i] * A[(r + 1) * F + j];
    const V a2 = A[(r + 2) * F + i] * A[(r + 2) * F + j];
    const V a3 = A[(r + 3) * F + i] * A[(r + 3) * F + j];
    acc += (a0 + a1) + (a2 + a3);
  }
  for (; r < r1; ++r) acc += A[r * F + i] * A[r * F + j];
  Gpart[(int64_t)blockIdx.x * F * F + e] = acc;
}

template <typename V>
__global__ void __launch_bounds__(256)
gram_det_fold_kern(const V * __restrict__ Gpart, int64_t nparts, int F,
                   V * __restrict__ G) {
  const int e = threadIdx.x;
  if (e >= F * F) return;
  V acc = (V)0;
  for (int64_t b = 0; b < nparts; ++b) acc += Gpart[b * F * F + e];
  G[e] = acc;
}

template <typename V>
int launch_rowsolve(const V * A, const V * B, V * C, int64_t n, int F,
                    hipStream_t st) {
  const int rows_pb = 256 / (F < 256 ? F : 256);
  int64_t blocks = (n + rows_pb - 1) / rows_pb;
  if (blocks > 8192) blocks = 8192;          // grid-stride beyond this
  if (blocks < 1) blocks = 1;
  dim3 g((uint32_t)blocks), b(256);
  switch (F) {
    case 4:  hipLaunchKernelGGL((rowsolve_kern<V, 4>),  g, b, 0, st, A, B, C, n); return 0;
    case 8:  hipLaunchKernelGGL((rowsolve_kern<V, 8>),  g, b, 0, st, A, B, C, n); return 0;
    case 16: hipLaunchKernelGGL((rowsolve_kern<V, 16>), g, b, 0, st, A, B, C, n); return 0;
    case 32: hipLaunchKernelGGL((rowsolve_kern<V, 32>), g, b, 0, st, A, B, C, n); return 0;
    case 64: hipLaunchKernelGGL((rowsolve_kern<V, 64>), g, b, 0, st, A, B, C, n); return 0;
    default: return -1;
  }
}

}  // namespace

extern "C" void splatt_hip_gram_det_f64(const double * A, int64_t n, int F,
                                        double * Gpart, int64_t nparts,
                                        double * G, void * stream) {
  const int64_t rpb = (n + nparts - 1) / nparts;
  hipLaunchKernelGGL((gram_det_part_kern<double>), dim3((uint32_t)nparts),
                     dim3(256), 0, (hipStream_t)stream, A, n, F, rpb, Gpart);
  hipLaunchKernelGGL((gram_det_fold_kern<double>), dim3(1), dim3(256), 0,
                     (hipStream_t)stream, Gpart, nparts, F, G);
}
extern "C" void splatt_hip_gram_det_f32(const float * A, int64_t n, int F,
                                        float * Gpart, int64_t nparts,
                                        float * G, void * stream) {
  const int64_t rpb = (n + nparts - 1) / nparts;
  hipLaunchKernelGGL((gram_det_part_kern<float>), dim3((uint32_t)nparts),
                     dim3(256), 0, (hipStream_t)stream, A, n, F, rpb, Gpart);
  hipLaunchKernelGGL((gram_det_fold_kern<float>), dim3(1), dim3(256), 0,
                     (hipStream_t)stream, Gpart, nparts, F, G);
}

// ---- generic column reductions/scales for the torch-free C API device
// driver (csrc/capi/capi_gpu.cpp): F <= 64, one lane group per column.
namespace {

__device__ inline void atomic_max_f64(double * p, double v) {
  unsigned long long * u = reinterpret_cast<unsigned long long *>(p);
  unsigned long long old = *u, assumed;
  while (__longlong_as_double(old) < v) {
    assumed = old;
    old = atomicCAS(u, assumed, __double_as_longlong(v));
    if (old == assumed) break;
  }
}

// which: 0 = sum of squares, 1 = max(|x|); out[F] pre-initialized
// (0 for sum, 0 for max — values are magnitudes)
__global__ void __launch_bounds__(256)
colacc_kern(const double * __restrict__ A, int64_t n, int F, int which,
            double * __restrict__ out) {
  const int gpc = (int)(blockDim.x / F);      // row groups per block
  const int tid = (int)threadIdx.x;
  if (tid >= gpc * F) return;
  const int c = tid % F;
  const int64_t g = (int64_t)blockIdx.x * gpc + tid / F;
  const int64_t stride = (int64_t)gridDim.x * gpc;
  double acc = 0.0;
  for (int64_t i = g; i < n; i += stride) {
    const double x = A[i * F + c];
    if (which == 0) acc += x * x;
    else acc = fmax(acc, fabs(x));
  }
  if (which == 0) atomic_add_g(&out[c], acc);
  else atomic_max_f64(&out[c], acc);
}

__global__ void __launch_bounds__(256)
colscale_kern(double * __restrict__ A, int64_t n, int F,
              const double * __restrict__ lam) {
  const int64_t t0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t e = t0; e < n * F; e += stride)
    A[e] /= lam[e % F];
}

// out[f] += sum_i X[i,f] * A[i,f]  (fit inner product; out pre-zeroed)
__global__ void __launch_bounds__(256)
coldot_kern(const double * __restrict__ X, const double * __restrict__ A,
            int64_t n, int F, double * __restrict__ out) {
  const int gpc = (int)(blockDim.x / F);
  const int tid = (int)threadIdx.x;
  if (tid >= gpc * F) return;
  const int c = tid % F;
  const int64_t g = (int64_t)blockIdx.x * gpc + tid / F;
  const int64_t stride = (int64_t)gridDim.x * gpc;
  double acc = 0.0;
  for (int64_t i = g; i < n; i += stride)
    acc += X[i * F + c] * A[i * F + c];
  atomic_add_g(&out[c], acc);
}

}  // namespace

extern "C" void splatt_hip_colacc_f64(const double * A, int64_t n, int F,
                                      int which, double * out,
                                      void * stream) {
  hipLaunchKernelGGL(colacc_kern, dim3(256), dim3(256), 0,
                     (hipStream_t)stream, A, n, F, which, out);
}
extern "C" void splatt_hip_colscale_f64(double * A, int64_t n, int F,
                                        const double * lam, void * stream) {
  hipLaunchKernelGGL(colscale_kern, dim3(512), dim3(256), 0,
                     (hipStream_t)stream, A, n, F, lam);
}
extern "C" void splatt_hip_coldot_f64(const double * X, const double * A,
                                      int64_t n, int F, double * out,
                                      void * stream) {
  hipLaunchKernelGGL(coldot_kern, dim3(256), dim3(256), 0,
                     (hipStream_t)stream, X, A, n, F, out);
}

extern "C" int splatt_hip_rowsolve_f64(const double * A, const double * B,
                                       double * C, int64_t n, int F,
                                       void * stream) {
  return launch_rowsolve<double>(A, B, C, n, F, (hipStream_t)stream);
}
extern "C" int splatt_hip_rowsolve_f32(const float * A, const float * B,
                                       float * C, int64_t n, int F,
                                       void * stream) {
  return launch_rowsolve<float>(A, B, C, n, F, (hipStream_t)stream);
}

extern "C" void splatt_hip_gram_f64(const double * A, int64_t n, int F,
                                    double * G, void * stream) {
  launch_gram<double>(A, n, F, G, (hipStream_t)stream);
}
extern "C" void splatt_hip_gram_f32(const float * A, int64_t n, int F,
                                    float * G, void * stream) {
  launch_gram<float>(A, n, F, G, (hipStream_t)stream);
}
extern "C" void splatt_hip_spd_inverse_f64(const double * G, double * Ginv,
                                           int F, void * stream) {
  hipLaunchKernelGGL((spd_inverse_kern<double>), dim3(1), dim3(64), 0,
                     (hipStream_t)stream, G, Ginv, F);
}
extern "C" void splatt_hip_spd_inverse_f32(const float * G, float * Ginv,
                                           int F, void * stream) {
  hipLaunchKernelGGL((spd_inverse_kern<float>), dim3(1), dim3(64), 0,
                     (hipStream_t)stream, G, Ginv, F);
}
