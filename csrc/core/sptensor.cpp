#include "sptensor.hpp"
#include <algorithm>
#include <numeric>
#include <cstdlib>
#include <cstring>
#if defined(_OPENMP)
#include <parallel/algorithm>
#include <omp.h>
#endif

namespace splatt {

void * aligned_alloc64(size_t bytes) {
  void * p = nullptr;
  if (posix_memalign(&p, 64, bytes) != 0) throw std::bad_alloc();
  return p;
}
void aligned_free64(void * ptr) { free(ptr); }

template <typename V>
void coo_sort(SpTensor<V> & tt, const int * perm) {
  const int nm = tt.nmodes;
  const idx_t nnz = tt.nnz;
  std::vector<uint64_t> order(nnz);
  std::iota(order.begin(), order.end(), (uint64_t)0);

  const idx_t * col[MAX_NMODES];
  for (int m = 0; m < nm; ++m) col[m] = tt.ind[perm[m]].data();

  auto cmp = [&](uint64_t a, uint64_t b) {
    for (int m = 0; m < nm; ++m) {
      const idx_t xa = col[m][a], xb = col[m][b];
      if (xa != xb) return xa < xb;
    }
    return a < b;  // stable tie-break: duplicates keep input order
  };
#if defined(_OPENMP)
  __gnu_parallel::sort(order.begin(), order.end(), cmp);
#else
  std::sort(order.begin(), order.end(), cmp);
#endif

  // gather into sorted order
  std::vector<idx_t> tmp(nnz);
  for (int m = 0; m < nm; ++m) {
    idx_t * src = tt.ind[m].data();
    #pragma omp parallel for schedule(static)
    for (int64_t i = 0; i < (int64_t)nnz; ++i) tmp[i] = src[order[i]];
    std::memcpy(src, tmp.data(), nnz * sizeof(idx_t));
  }
  std::vector<V> vtmp(nnz);
  #pragma omp parallel for schedule(static)
  for (int64_t i = 0; i < (int64_t)nnz; ++i) vtmp[i] = tt.vals[order[i]];
  tt.vals.swap(vtmp);
}

template <typename V>
idx_t coo_remove_dups(SpTensor<V> & tt) {
  const int nm = tt.nmodes;
  if (tt.nnz == 0) return 0;
  idx_t w = 0;
  for (idx_t r = 1; r < tt.nnz; ++r) {
    bool same = true;
    for (int m = 0; m < nm; ++m) same &= (tt.ind[m][r] == tt.ind[m][w]);
    if (same) {
      tt.vals[w] += tt.vals[r];
    } else {
      ++w;
      for (int m = 0; m < nm; ++m) tt.ind[m][w] = tt.ind[m][r];
      tt.vals[w] = tt.vals[r];
    }
  }
  const idx_t removed = tt.nnz - (w + 1);
  tt.nnz = w + 1;
  for (int m = 0; m < nm; ++m) tt.ind[m].resize(tt.nnz);
  tt.vals.resize(tt.nnz);
  return removed;
}

template <typename V>
idx_t coo_remove_empty(SpTensor<V> & tt) {
  idx_t total_removed = 0;
  for (int m = 0; m < tt.nmodes; ++m) {
    std::vector<idx_t> hist = coo_hist(tt, m);
    const idx_t dim = tt.dims[m];
    idx_t nonempty = 0;
    for (idx_t s = 0; s < dim; ++s) nonempty += (hist[s] != 0);
    if (nonempty == dim) continue;

    std::vector<idx_t> relabel(dim);
    std::vector<idx_t> map(nonempty);
    idx_t nxt = 0;
    for (idx_t s = 0; s < dim; ++s) {
      if (hist[s] != 0) { relabel[s] = nxt; map[nxt] = s; ++nxt; }
    }
    idx_t * col = tt.ind[m].data();
    #pragma omp parallel for schedule(static)
    for (int64_t i = 0; i < (int64_t)tt.nnz; ++i) col[i] = relabel[col[i]];
    total_removed += dim - nonempty;
    tt.dims[m] = nonempty;
    tt.indmap[m] = std::move(map);
  }
  return total_removed;
}

template <typename V>
std::vector<idx_t> coo_hist(const SpTensor<V> & tt, int mode) {
  std::vector<idx_t> hist(tt.dims[mode], 0);
  const idx_t * col = tt.ind[mode].data();
  for (idx_t i = 0; i < tt.nnz; ++i) ++hist[col[i]];
  return hist;
}

template struct SpTensor<float>;
template struct SpTensor<double>;
template void coo_sort<float>(SpTensor<float>&, const int*);
template void coo_sort<double>(SpTensor<double>&, const int*);
template idx_t coo_remove_dups<float>(SpTensor<float>&);
template idx_t coo_remove_dups<double>(SpTensor<double>&);
template idx_t coo_remove_empty<float>(SpTensor<float>&);
template idx_t coo_remove_empty<double>(SpTensor<double>&);
template std::vector<idx_t> coo_hist<float>(const SpTensor<float>&, int);
template std::vector<idx_t> coo_hist<double>(const SpTensor<double>&, int);

}  // namespace splatt
