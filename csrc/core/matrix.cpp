#include "matrix.hpp"
#include <cmath>
#include <cstring>
#include <algorithm>
#if defined(_OPENMP)
#include <omp.h>
#endif

namespace splatt {

template <typename V>
void mat_ata(const V * A, idx_t n, int F, V * G) {
  std::memset(G, 0, sizeof(V) * F * F);
#if defined(_OPENMP)
  int nt = omp_get_max_threads();
#else
  int nt = 1;
#endif
  std::vector<double> partial((size_t)nt * F * F, 0.0);
  #pragma omp parallel
  {
#if defined(_OPENMP)
    const int t = omp_get_thread_num();
#else
    const int t = 0;
#endif
    double * P = partial.data() + (size_t)t * F * F;
    #pragma omp for schedule(static)
    for (int64_t i = 0; i < (int64_t)n; ++i) {
      const V * row = A + (idx_t)i * F;
      for (int a = 0; a < F; ++a) {
        const double ra = (double)row[a];
        double * Pr = P + (size_t)a * F;
        for (int b = a; b < F; ++b) Pr[b] += ra * (double)row[b];
      }
    }
  }
  for (int t = 0; t < nt; ++t) {
    const double * P = partial.data() + (size_t)t * F * F;
    for (int a = 0; a < F; ++a)
      for (int b = a; b < F; ++b) G[a * F + b] += (V)P[(size_t)a * F + b];
  }
  for (int a = 0; a < F; ++a)
    for (int b = 0; b < a; ++b) G[a * F + b] = G[b * F + a];
}

template <typename V>
void gram_hadamard(V const * const * grams, int nmats, int skip, int F,
                   V * G, V reg) {
  for (int i = 0; i < F * F; ++i) G[i] = (V)1;
  for (int m = 0; m < nmats; ++m) {
    if (m == skip) continue;
    const V * g = grams[m];
    for (int i = 0; i < F * F; ++i) G[i] *= g[i];
  }
  for (int f = 0; f < F; ++f) G[f * F + f] += reg;
}

namespace {

// Cholesky G = L L^T in place (lower). Returns false on non-positive pivot.
template <typename V>
bool cholesky(V * L, int F) {
  for (int j = 0; j < F; ++j) {
    double d = (double)L[j * F + j];
    for (int k = 0; k < j; ++k) d -= (double)L[j * F + k] * (double)L[j * F + k];
    if (!(d > 0)) return false;
    const double dj = std::sqrt(d);
    L[j * F + j] = (V)dj;
    for (int i = j + 1; i < F; ++i) {
      double s = (double)L[i * F + j];
      for (int k = 0; k < j; ++k) s -= (double)L[i * F + k] * (double)L[j * F + k];
      L[i * F + j] = (V)(s / dj);
    }
  }
  return true;
}

}  // namespace

template <typename V>
int solve_normals(V * B, idx_t n, int F, const V * G, V reg0) {
  std::vector<V> L((size_t)F * F);
  int bumps = 0;
  V reg = reg0;  // user ridge term (reference p_form_gram + reg*I)
  // escalate Tikhonov regularization until the Cholesky succeeds
  for (;;) {
    std::memcpy(L.data(), G, sizeof(V) * F * F);
    if (reg > (V)0)
      for (int f = 0; f < F; ++f) L[(size_t)f * F + f] += reg;
    if (cholesky(L.data(), F)) break;
    ++bumps;
    V scale = (V)0;
    for (int f = 0; f < F; ++f) scale = std::max(scale, std::abs(G[(size_t)f * F + f]));
    reg = (reg == (V)0) ? scale * (V)1e-12 : reg * (V)100;
    if (bumps > 20) break;
  }
  // Solve X G = B  ==>  G X^T = B^T; with G = L L^T, for each row b of B:
  // solve L y = b, then L^T x = y  (G symmetric).
  #pragma omp parallel for schedule(static)
  for (int64_t i = 0; i < (int64_t)n; ++i) {
    V * row = B + (idx_t)i * F;
    // forward: L y = row
    for (int a = 0; a < F; ++a) {
      double s = (double)row[a];
      for (int k = 0; k < a; ++k) s -= (double)L[(size_t)a * F + k] * (double)row[k];
      row[a] = (V)(s / (double)L[(size_t)a * F + a]);
    }
    // backward: L^T x = y
    for (int a = F - 1; a >= 0; --a) {
      double s = (double)row[a];
      for (int k = a + 1; k < F; ++k) s -= (double)L[(size_t)k * F + a] * (double)row[k];
      row[a] = (V)(s / (double)L[(size_t)a * F + a]);
    }
  }
  return bumps;
}

template <typename V>
void mat_normalize(V * A, idx_t n, int F, V * lambda, int which) {
  std::vector<double> acc(F, 0.0);
  if (which == 0) {
    #pragma omp parallel
    {
      std::vector<double> loc(F, 0.0);
      #pragma omp for schedule(static)
      for (int64_t i = 0; i < (int64_t)n; ++i) {
        const V * row = A + (idx_t)i * F;
        for (int f = 0; f < F; ++f) loc[f] += (double)row[f] * (double)row[f];
      }
      #pragma omp critical
      for (int f = 0; f < F; ++f) acc[f] += loc[f];
    }
    for (int f = 0; f < F; ++f) lambda[f] = (V)std::sqrt(acc[f]);
  } else {
    #pragma omp parallel
    {
      std::vector<double> loc(F, 0.0);
      #pragma omp for schedule(static)
      for (int64_t i = 0; i < (int64_t)n; ++i) {
        const V * row = A + (idx_t)i * F;
        for (int f = 0; f < F; ++f) loc[f] = std::max(loc[f], (double)std::abs(row[f]));
      }
      #pragma omp critical
      for (int f = 0; f < F; ++f) acc[f] = std::max(acc[f], loc[f]);
    }
    // max-norm clamps at 1 so factors only shrink (reference semantics)
    for (int f = 0; f < F; ++f) lambda[f] = (V)std::max(acc[f], 1.0);
  }
  for (int f = 0; f < F; ++f) if (lambda[f] == (V)0) lambda[f] = (V)1;
  #pragma omp parallel for schedule(static)
  for (int64_t i = 0; i < (int64_t)n; ++i) {
    V * row = A + (idx_t)i * F;
    for (int f = 0; f < F; ++f) row[f] /= lambda[f];
  }
}

#define INST(V) \
  template void mat_ata<V>(const V*, idx_t, int, V*); \
  template void gram_hadamard<V>(V const* const*, int, int, int, V*, V); \
  template int solve_normals<V>(V*, idx_t, int, const V*, V); \
  template void mat_normalize<V>(V*, idx_t, int, V*, int);
INST(float)
INST(double)
#undef INST

}  // namespace splatt
