// Dense factor-matrix kernels for the ALS normal equations.
// Capability parity: reference src/matrix.c (mat_aTa:414, p_form_gram:29,
// mat_solve_normals:529 with potrf/potrs + SVD fallback, mat_normalize:501).
// Fresh design: self-contained blocked C++ (no external BLAS/LAPACK
// dependency for the host path — rank F <= 512 keeps the F x F work tiny);
// the GPU path uses rocBLAS via PyTorch instead.
#pragma once

#include "types.hpp"

namespace splatt {

// G += A^T A for row-major A (n x F). G is F x F row-major (zeroed inside).
template <typename V>
void mat_ata(const V * A, idx_t n, int F, V * G);

// G = hadamard of all grams except `skip`, + reg * I
template <typename V>
void gram_hadamard(V const * const * grams, int nmats, int skip, int F,
                   V * G, V reg = (V)0);

// Solve X * G = B for X (n x F), G SPD (F x F). B is overwritten with X.
// Cholesky; on breakdown retries with escalating diagonal regularization
// (the reference falls back to gelss SVD — we use Tikhonov instead and
// report it via the return value: 0 = clean, k = #bumps applied).
template <typename V>
int solve_normals(V * B, idx_t n, int F, const V * G,
                  V reg0 = (V)0);

// Column 2-norms (or max-norms) of A (n x F) into lambda, then scale columns
// to unit norm. which: 0 = 2-norm, 1 = max-norm.
template <typename V>
void mat_normalize(V * A, idx_t n, int F, V * lambda, int which);

}  // namespace splatt
