#include "csf.hpp"
#include <algorithm>
#include <numeric>
#include <cstring>
#if defined(_OPENMP)
#include <omp.h>
#endif

namespace splatt {

void order_smallfirst(const idx_t * dims, int nmodes, int * perm) {
  std::iota(perm, perm + nmodes, 0);
  std::stable_sort(perm, perm + nmodes,
                   [&](int a, int b) { return dims[a] < dims[b]; });
}

void order_root(const idx_t * dims, int nmodes, int mode, int * perm) {
  order_smallfirst(dims, nmodes, perm);
  // move `mode` to front, keep relative order of the rest
  int pos = 0;
  while (perm[pos] != mode) ++pos;
  for (int i = pos; i > 0; --i) perm[i] = perm[i - 1];
  perm[0] = mode;
}

void order_leaf(const idx_t * dims, int nmodes, int mode, int * perm) {
  order_smallfirst(dims, nmodes, perm);
  int pos = 0;
  while (perm[pos] != mode) ++pos;
  for (int i = pos; i < nmodes - 1; ++i) perm[i] = perm[i + 1];
  perm[nmodes - 1] = mode;
}

namespace {

// diff[i] = shallowest level at which sorted nonzero i differs from i-1
// (diff[0] = 0). A level-l node starts at every i with diff[i] <= l.
template <typename V>
std::vector<int8_t> p_first_diff_level(const SpTensor<V> & tt, const int * perm) {
  const int nm = tt.nmodes;
  const idx_t nnz = tt.nnz;
  std::vector<int8_t> diff(nnz);
  const idx_t * col[MAX_NMODES];
  for (int l = 0; l < nm; ++l) col[l] = tt.ind[perm[l]].data();
  diff[0] = 0;
  #pragma omp parallel for schedule(static)
  for (int64_t i = 1; i < (int64_t)nnz; ++i) {
    int8_t d = (int8_t)(nm - 1);  // always a new leaf node
    for (int l = 0; l < nm - 1; ++l) {
      if (col[l][i] != col[l][i - 1]) { d = (int8_t)l; break; }
    }
    diff[i] = d;
  }
  return diff;
}

// positions i in [0,nnz) with diff[i] <= level, in order (parallel filter)
std::vector<int64_t> p_filter_starts(const std::vector<int8_t> & diff, int level) {
  const int64_t n = (int64_t)diff.size();
  int nchunks = 1;
#if defined(_OPENMP)
  nchunks = std::max(1, omp_get_max_threads() * 4);
#endif
  const int64_t chunk = (n + nchunks - 1) / nchunks;
  std::vector<int64_t> counts(nchunks + 1, 0);
  #pragma omp parallel for schedule(static)
  for (int c = 0; c < nchunks; ++c) {
    const int64_t lo = c * chunk, hi = std::min(n, lo + chunk);
    int64_t cnt = 0;
    for (int64_t i = lo; i < hi; ++i) cnt += (diff[i] <= level);
    counts[c + 1] = cnt;
  }
  for (int c = 0; c < nchunks; ++c) counts[c + 1] += counts[c];
  std::vector<int64_t> out(counts[nchunks]);
  #pragma omp parallel for schedule(static)
  for (int c = 0; c < nchunks; ++c) {
    const int64_t lo = c * chunk, hi = std::min(n, lo + chunk);
    int64_t w = counts[c];
    for (int64_t i = lo; i < hi; ++i)
      if (diff[i] <= level) out[w++] = i;
  }
  return out;
}

}  // namespace

template <typename V>
Csf<V> csf_build(SpTensor<V> & tt, const int * perm) {
  Csf<V> c;
  const int nm = tt.nmodes;
  for (int m = 0; m < nm; ++m)
    if (tt.dims[m] > 0xFFFFFFFFull)
      throw std::runtime_error(
          "mode dimension above 2^32 (32-bit CSF node ids)");
  c.nmodes = nm;
  c.nnz = tt.nnz;
  for (int m = 0; m < nm; ++m) c.dims[m] = tt.dims[m];
  for (int l = 0; l < nm; ++l) { c.dim_perm[l] = perm[l]; c.dim_iperm[perm[l]] = l; }

  coo_sort(tt, perm);
  if (tt.nnz == 0) return c;

  const auto diff = p_first_diff_level(tt, perm);

  // per-level start positions (starts[nm-1] is implicit: every nnz)
  std::array<std::vector<int64_t>, MAX_NMODES> starts;
  for (int l = 0; l < nm - 1; ++l) starts[l] = p_filter_starts(diff, l);

  // leaf level: fids = sorted leaf indices, vals copied
  c.nfibs[nm - 1] = tt.nnz;
  c.fids[nm - 1].resize(tt.nnz);
  c.vals.resize(tt.nnz);
  {
    const idx_t * leafcol = tt.ind[perm[nm - 1]].data();
    #pragma omp parallel for schedule(static)
    for (int64_t i = 0; i < (int64_t)tt.nnz; ++i) {
      c.fids[nm - 1][i] = (fid_t)leafcol[i];
      c.vals[i] = tt.vals[i];
    }
  }

  for (int l = 0; l < nm - 1; ++l) {
    const int64_t nf = (int64_t)starts[l].size();
    c.nfibs[l] = (idx_t)nf;
    c.fids[l].resize(nf);
    c.fptr[l].resize(nf + 1);
    const idx_t * lcol = tt.ind[perm[l]].data();
    const bool leaf_child = (l == nm - 2);
    const std::vector<int64_t> * child = leaf_child ? nullptr : &starts[l + 1];
    #pragma omp parallel for schedule(static)
    for (int64_t k = 0; k < nf; ++k) {
      const int64_t pos = starts[l][k];
      c.fids[l][k] = (fid_t)lcol[pos];
      // child range start = rank of pos among level-(l+1) starts
      if (leaf_child) {
        c.fptr[l][k] = pos;
      } else {
        c.fptr[l][k] = std::lower_bound(child->begin(), child->end(), pos)
                       - child->begin();
      }
    }
    c.fptr[l][nf] = leaf_child ? (int64_t)tt.nnz
                               : (int64_t)starts[l + 1].size();
  }

  // dense root => drop root fids (identity), matching reference semantics
  if (c.nfibs[0] == c.dims[perm[0]]) c.fids[0].clear();
  return c;
}

template <typename V>
CsfSet<V> csf_alloc(SpTensor<V> & tt, const Options & opts) {
  CsfSet<V> set;
  const int nm = tt.nmodes;
  int perm[MAX_NMODES];

  switch (opts.csf_alloc) {
    case CsfAlloc::ONEMODE: {
      order_smallfirst(tt.dims.data(), nm, perm);
      set.csfs.push_back(csf_build(tt, perm));
      for (int m = 0; m < nm; ++m) {
        set.mode_csf[m] = 0;
        set.mode_depth[m] = set.csfs[0].level_of_mode(m);
      }
      break;
    }
    case CsfAlloc::TWOMODE: {
      order_smallfirst(tt.dims.data(), nm, perm);
      const int longest = perm[nm - 1];
      set.csfs.push_back(csf_build(tt, perm));
      // second copy rooted at the longest mode -> lock-free root kernel there
      order_root(tt.dims.data(), nm, longest, perm);
      set.csfs.push_back(csf_build(tt, perm));
      for (int m = 0; m < nm; ++m) {
        if (m == longest) { set.mode_csf[m] = 1; set.mode_depth[m] = 0; }
        else { set.mode_csf[m] = 0; set.mode_depth[m] = set.csfs[0].level_of_mode(m); }
      }
      break;
    }
    case CsfAlloc::ALLMODE: {
      for (int m = 0; m < nm; ++m) {
        order_root(tt.dims.data(), nm, m, perm);
        set.csfs.push_back(csf_build(tt, perm));
        set.mode_csf[m] = m;
        set.mode_depth[m] = 0;
      }
      break;
    }
  }
  return set;
}

template <typename V>
double csf_frobsq(const Csf<V> & c) {
  double acc = 0;
  #pragma omp parallel for schedule(static) reduction(+:acc)
  for (int64_t i = 0; i < (int64_t)c.vals.size(); ++i)
    acc += (double)c.vals[i] * (double)c.vals[i];
  return acc;
}

template struct Csf<float>;
template struct Csf<double>;
template Csf<float> csf_build<float>(SpTensor<float>&, const int*);
template Csf<double> csf_build<double>(SpTensor<double>&, const int*);
template CsfSet<float> csf_alloc<float>(SpTensor<float>&, const Options&);
template CsfSet<double> csf_alloc<double>(SpTensor<double>&, const Options&);
template double csf_frobsq<float>(const Csf<float>&);
template double csf_frobsq<double>(const Csf<double>&);

}  // namespace splatt
