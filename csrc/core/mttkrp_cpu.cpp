#include "mttkrp_cpu.hpp"
#include "partition.hpp"
#include <cstring>
#if defined(_OPENMP)
#include <omp.h>
#endif

namespace splatt {

template <typename V>
void mttkrp_stream(const SpTensor<V> & tt, V const * const * mats,
                   V * out, int mode, int rank) {
  const int nm = tt.nmodes;
  std::memset(out, 0, sizeof(V) * tt.dims[mode] * rank);
  std::vector<V> accum(rank);
  for (idx_t x = 0; x < tt.nnz; ++x) {
    const V v = tt.vals[x];
    for (int f = 0; f < rank; ++f) accum[f] = v;
    for (int m = 0; m < nm; ++m) {
      if (m == mode) continue;
      const V * row = mats[m] + tt.ind[m][x] * rank;
      for (int f = 0; f < rank; ++f) accum[f] *= row[f];
    }
    V * orow = out + tt.ind[mode][x] * rank;
    for (int f = 0; f < rank; ++f) orow[f] += accum[f];
  }
}

namespace {

// Subtree accumulation for levels (level..leaf): returns into `buf` the
// F-vector g(node) = M_l[fid] (.) sum_children g(child), leaf g = v * M_leaf[fid].
// `include_self`: whether to multiply by this node's own factor row.
template <typename V>
struct CsfWalker {
  const Csf<V> & c;
  V const * const * mats;   // indexed by MODE
  V * out;
  int rank;
  int outdepth;
  int leaf;
  bool use_atomics = true;  // false when the output is thread-private

  // accumulate g over children of (level, node) into acc (acc zeroed by caller)
  void subtree_below(int level, int64_t node, V * acc, V * scratch) const {
    const int child_level = level + 1;
    const int64_t start = c.fptr[level][node];
    const int64_t end = c.fptr[level][node + 1];
    const V * M = mats[c.dim_perm[child_level]];
    if (child_level == leaf) {
      for (int64_t j = start; j < end; ++j) {
        const V v = c.vals[j];
        const V * row = M + (idx_t)c.fids[leaf][j] * rank;
        for (int f = 0; f < rank; ++f) acc[f] += v * row[f];
      }
    } else {
      for (int64_t n = start; n < end; ++n) {
        for (int f = 0; f < rank; ++f) scratch[f] = 0;
        subtree_below(child_level, n, scratch, scratch + rank);
        const V * row = M + (idx_t)c.fids[child_level][n] * rank;
        for (int f = 0; f < rank; ++f) acc[f] += scratch[f] * row[f];
      }
    }
  }

  // walk down maintaining the Hadamard product of ancestor rows (`above`),
  // emit at outdepth (intl/leaf output classes)
  void walk_down(int level, int64_t node, const V * above, V * bufs) const {
    const int64_t start = c.fptr[level][node];
    const int64_t end = c.fptr[level][node + 1];
    const int child_level = level + 1;
    const V * M = mats[c.dim_perm[child_level]];
    if (child_level == outdepth) {
      if (child_level == leaf) {
        // leaf output: out[fid] += v * above
        for (int64_t j = start; j < end; ++j) {
          const V v = c.vals[j];
          V * orow = out + (idx_t)c.fids[leaf][j] * rank;
          if (use_atomics) {
            for (int f = 0; f < rank; ++f) {
              #pragma omp atomic
              orow[f] += v * above[f];
            }
          } else {
            for (int f = 0; f < rank; ++f) orow[f] += v * above[f];
          }
        }
      } else {
        // internal output: out[fid] += above (.) below
        for (int64_t n = start; n < end; ++n) {
          V * below = bufs;
          for (int f = 0; f < rank; ++f) below[f] = 0;
          subtree_below(child_level, n, below, bufs + rank);
          V * orow = out + (idx_t)c.fids[child_level][n] * rank;
          if (use_atomics) {
            for (int f = 0; f < rank; ++f) {
              #pragma omp atomic
              orow[f] += above[f] * below[f];
            }
          } else {
            for (int f = 0; f < rank; ++f) orow[f] += above[f] * below[f];
          }
        }
      }
    } else {
      for (int64_t n = start; n < end; ++n) {
        V * nxt = bufs;
        const V * row = M + (idx_t)c.fids[child_level][n] * rank;
        for (int f = 0; f < rank; ++f) nxt[f] = above[f] * row[f];
        walk_down(child_level, n, nxt, bufs + rank);
      }
    }
  }
};

}  // namespace

template <typename V>
void mttkrp_csf_cpu(const Csf<V> & c, V const * const * mats,
                    V * out, int mode, int rank, int nthreads) {
  const int nm = c.nmodes;
  const int outdepth = c.level_of_mode(mode);
  const int leaf = nm - 1;
  std::memset(out, 0, sizeof(V) * c.dims[mode] * rank);
  if (c.nnz == 0) return;

  const bool root_labeled = !c.fids[0].empty();
  const int64_t nroot = (int64_t)c.nfibs[0];

#if defined(_OPENMP)
  if (nthreads > 0) omp_set_num_threads(nthreads);
#endif

  CsfWalker<V> w{c, mats, out, rank, outdepth, leaf, true};

  if (outdepth == 0) {
    // root output: no write conflicts across root nodes. Load balance via
    // CCP over per-root-subtree nnz (reference csf_partition_1d,
    // csf.c:854-872) instead of dynamic chunking.
    std::vector<int64_t> weights(nroot);
    {
      // nnz span per root node: compose fptr chains to the leaf level
      std::vector<int64_t> nnzstart(c.fptr[0]);
      for (int l = 1; l < nm - 1; ++l) {
        #pragma omp parallel for schedule(static)
        for (int64_t k = 0; k < (int64_t)nnzstart.size(); ++k)
          nnzstart[k] = c.fptr[l][nnzstart[k]];
      }
      #pragma omp parallel for schedule(static)
      for (int64_t s = 0; s < nroot; ++s)
        weights[s] = nnzstart[s + 1] - nnzstart[s];
    }
    int nt = 1;
#if defined(_OPENMP)
    nt = omp_get_max_threads();
#endif
    const auto parts = partition_weighted(weights.data(), nroot, nt);
    #pragma omp parallel
    {
#if defined(_OPENMP)
      const int tid = omp_get_thread_num();
#else
      const int tid = 0;
#endif
      std::vector<V> bufs((size_t)rank * (nm + 1));
      for (int64_t s = parts[tid]; s < parts[tid + 1]; ++s) {
        V * acc = bufs.data();
        for (int f = 0; f < rank; ++f) acc[f] = 0;
        w.subtree_below(0, s, acc, bufs.data() + rank);
        const idx_t orow_i = root_labeled ? (idx_t)c.fids[0][s] : (idx_t)s;
        V * orow = out + orow_i * rank;
        for (int f = 0; f < rank; ++f) orow[f] += acc[f];
      }
    }
  } else {
    const V * Mroot = mats[c.dim_perm[0]];
    int nt = 1;
#if defined(_OPENMP)
    nt = omp_get_max_threads();
#endif
    // privatization heuristic (reference p_is_privatized, mttkrp.c:221-236):
    // replicate the output per thread when it is small relative to nnz,
    // then tree-reduce — avoids the per-element atomics of the scatter
    // walkers for short modes.
    const size_t out_elems = (size_t)c.dims[mode] * rank;
    const bool privatize =
        nt > 1 && (double)out_elems * nt <= 0.1 * (double)c.nnz * rank;
    if (privatize) {
      std::vector<V> priv((size_t)nt * out_elems, (V)0);
      #pragma omp parallel
      {
#if defined(_OPENMP)
        const int tid = omp_get_thread_num();
#else
        const int tid = 0;
#endif
        CsfWalker<V> wp{c, mats, priv.data() + (size_t)tid * out_elems,
                        rank, outdepth, leaf, /*use_atomics=*/false};
        std::vector<V> bufs((size_t)rank * (nm + 2));
        #pragma omp for schedule(dynamic, 16)
        for (int64_t s = 0; s < nroot; ++s) {
          const idx_t rid = root_labeled ? (idx_t)c.fids[0][s] : (idx_t)s;
          wp.walk_down(0, s, Mroot + rid * rank, bufs.data());
        }
        // tree-style reduction: each thread sums its block across copies
        #pragma omp barrier
        #pragma omp for schedule(static)
        for (int64_t e = 0; e < (int64_t)out_elems; ++e) {
          V acc = (V)0;
          for (int t = 0; t < nt; ++t) acc += priv[(size_t)t * out_elems + e];
          out[e] = acc;
        }
      }
    } else {
      #pragma omp parallel
      {
        std::vector<V> bufs((size_t)rank * (nm + 2));
        #pragma omp for schedule(dynamic, 16)
        for (int64_t s = 0; s < nroot; ++s) {
          const idx_t rid = root_labeled ? (idx_t)c.fids[0][s] : (idx_t)s;
          const V * above0 = Mroot + rid * rank;
          w.walk_down(0, s, above0, bufs.data());
        }
      }
    }
  }
}

template void mttkrp_stream<float>(const SpTensor<float>&, float const* const*, float*, int, int);
template void mttkrp_stream<double>(const SpTensor<double>&, double const* const*, double*, int, int);
template void mttkrp_csf_cpu<float>(const Csf<float>&, float const* const*, float*, int, int, int);
template void mttkrp_csf_cpu<double>(const Csf<double>&, double const* const*, double*, int, int, int);

}  // namespace splatt
