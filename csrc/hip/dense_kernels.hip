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

template <typename V>
void launch_gram(const V * A, int64_t n, int F, V * G, hipStream_t st) {
  // enough blocks to fill the chip, >= 4 LDS tiles of work each
  int64_t nblocks = (n + TR * 4 - 1) / (TR * 4);
  if (nblocks > 2048) nblocks = 2048;
  if (nblocks < 1) nblocks = 1;
  const int64_t rows_per_blk = (n + nblocks - 1) / nblocks;
  dim3 grid((uint32_t)nblocks), block(TPB);
  const int ff = F * F;
  if (ff <= 256)
    hipLaunchKernelGGL((gram_kern<V, 1>), grid, block, 0, st, A, n, F, rows_per_blk, G);
  else if (ff <= 1024)
    hipLaunchKernelGGL((gram_kern<V, 4>), grid, block, 0, st, A, n, F, rows_per_blk, G);
  else
    hipLaunchKernelGGL((gram_kern<V, 16>), grid, block, 0, st, A, n, F, rows_per_blk, G);
}

}  // namespace

extern "C" void splatt_hip_gram_f64(const double * A, int64_t n, int F,
                                    double * G, void * stream) {
  launch_gram<double>(A, n, F, G, (hipStream_t)stream);
}
extern "C" void splatt_hip_gram_f32(const float * A, int64_t n, int F,
                                    float * G, void * stream) {
  launch_gram<float>(A, n, F, G, (hipStream_t)stream);
}
