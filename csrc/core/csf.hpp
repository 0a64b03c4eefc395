// CSF (Compressed Sparse Fiber) tensor — the central format.
// Capability parity: reference src/csf.{h,c} + include/splatt/structs.h:45-130.
// Fresh design for GPU residency: each CSF is a set of flat arrays per level
//   fptr[l] : int64 [nfibs[l]+1]  children ranges into level l+1 (l < leaf)
//   fids[l] : uint32 [nfibs[l]]   node labels (empty at root => identity)
//   vals    : V [nnz]             aligned with leaf level
// so the whole structure uploads to HBM as nmodes*2+1 contiguous buffers —
// no per-tile struct-of-pointers indirection like the reference's csf_sparsity.
#pragma once

#include "types.hpp"
#include "sptensor.hpp"

namespace splatt {

template <typename V>
struct Csf {
  int nmodes = 0;
  idx_t nnz = 0;
  std::array<idx_t, MAX_NMODES> dims{};      // tensor dims (mode order)
  std::array<int, MAX_NMODES> dim_perm{};    // level -> mode
  std::array<int, MAX_NMODES> dim_iperm{};   // mode -> level
  std::array<idx_t, MAX_NMODES> nfibs{};
  std::array<std::vector<int64_t>, MAX_NMODES> fptr;
  std::array<std::vector<fid_t>, MAX_NMODES> fids;
  std::vector<V> vals;

  int mode_at_level(int l) const { return dim_perm[l]; }
  int level_of_mode(int m) const { return dim_iperm[m]; }

  size_t storage_bytes() const {
    size_t b = vals.size() * sizeof(V);
    for (int l = 0; l < nmodes; ++l)
      b += fptr[l].size() * sizeof(int64_t) + fids[l].size() * sizeof(fid_t);
    return b;
  }
};

// Mode-order policies (reference: csf_find_mode_order, csf.c:694-726).
// smallfirst: modes sorted by increasing dimension (root = smallest).
// inner(m):   mode m at root, remaining sorted by increasing dimension.
// leaf(m):    mode m at leaf, remaining sorted by increasing dimension.
void order_smallfirst(const idx_t * dims, int nmodes, int * perm);
void order_root(const idx_t * dims, int nmodes, int mode, int * perm);
void order_leaf(const idx_t * dims, int nmodes, int mode, int * perm);

// Build one CSF with the given level->mode permutation. Sorts `tt` in place.
template <typename V>
Csf<V> csf_build(SpTensor<V> & tt, const int * perm);

// Allocation policies producing 1 / 2 / nmodes CSF copies and the
// mode->csf + mode->outdepth dispatch map (reference csf_alloc, csf.c:770-814).
template <typename V>
struct CsfSet {
  std::vector<Csf<V>> csfs;
  std::array<int, MAX_NMODES> mode_csf{};    // which csf to use for output mode m
  std::array<int, MAX_NMODES> mode_depth{};  // output depth in that csf
};

template <typename V>
CsfSet<V> csf_alloc(SpTensor<V> & tt, const Options & opts);

// Frobenius norm^2 of the tensor, double accumulation (csf.c:828-851).
template <typename V>
double csf_frobsq(const Csf<V> & c);

extern template struct Csf<float>;
extern template struct Csf<double>;

}  // namespace splatt
