// COO sparse tensor (host). Capability parity: reference src/sptensor.{h,c}
// (sptensor_t struct sptensor.h:27-41; dedup/remove-empty sptensor.c:135-229).
// Fresh design: SoA std::vectors, templated value type.
#pragma once

#include "types.hpp"
#include <cmath>

namespace splatt {

template <typename V>
struct SpTensor {
  int nmodes = 0;
  idx_t nnz = 0;
  std::array<idx_t, MAX_NMODES> dims{};
  std::array<std::vector<idx_t>, MAX_NMODES> ind;  // ind[m][n]
  std::vector<V> vals;
  // local->global index map after empty-slice compression (empty => identity)
  std::array<std::vector<idx_t>, MAX_NMODES> indmap;

  SpTensor() = default;
  SpTensor(int nm, idx_t nz, const idx_t * d) : nmodes(nm), nnz(nz) {
    for (int m = 0; m < nm; ++m) { dims[m] = d[m]; ind[m].resize(nz); }
    vals.resize(nz);
  }

  double normsq() const {
    double acc = 0;
    for (idx_t i = 0; i < nnz; ++i) acc += (double)vals[i] * (double)vals[i];
    return acc;
  }
  double density() const {
    double d = (double)nnz;
    for (int m = 0; m < nmodes; ++m) d /= (double)dims[m];
    return d;
  }
};

// Sort nonzeros lexicographically with mode `perm[0]` most significant.
// Parallel (OpenMP gnu parallel sort). Reference behavior: tt_sort
// (src/sort.c:912-961) — ours is a permutation-index sort + gather.
template <typename V>
void coo_sort(SpTensor<V> & tt, const int * perm);

// Sum duplicate entries (requires any full sort first); returns #removed.
// Reference: tt_remove_dups (sptensor.c:135-162).
template <typename V>
idx_t coo_remove_dups(SpTensor<V> & tt);

// Relabel each mode to remove empty slices; fills indmap (local->global).
// Reference: tt_remove_empty (sptensor.c:164-229).
template <typename V>
idx_t coo_remove_empty(SpTensor<V> & tt);

// Per-slice nonzero histogram for a mode.
template <typename V>
std::vector<idx_t> coo_hist(const SpTensor<V> & tt, int mode);

extern template struct SpTensor<float>;
extern template struct SpTensor<double>;

}  // namespace splatt
