// CPD-ALS driver (host path). Capability parity: reference src/cpd.c
// (splatt_cpd_als:22, cpd_als_iterate:271-387, fit math:116-265,
// post-process:391-411). The GPU driver lives in Python on top of the HIP
// MTTKRP; this host implementation is the CPU-reference config and the
// numerical oracle for it.
#pragma once

#include "csf.hpp"
#include "mttkrp_cpu.hpp"
#include "matrix.hpp"
#include <functional>

namespace splatt {

template <typename V>
struct Kruskal {
  int nmodes = 0;
  int rank = 0;
  std::array<idx_t, MAX_NMODES> dims{};
  std::array<std::vector<V>, MAX_NMODES> factors;  // row-major dims[m] x rank
  std::vector<V> lambda;
  double fit = 0;
  int niters = 0;
};

// Deterministic seeded factor init, uniform [0,1): element (i,f) of mode m
// depends only on (seed, m, i, f) — rank-count and partition invariant
// (the property the reference gets from root-generated mpi_mat_rand,
// mpi/mpi_io.c:1097-1176).
template <typename V>
void seeded_factor_init(V * A, idx_t nrows, int rank, idx_t row0,
                        uint64_t seed, int mode);

template <typename V>
Kruskal<V> cpd_als(const CsfSet<V> & set, int rank, const Options & opts,
                   std::function<void(int, double, double, double)> iter_cb = {});

}  // namespace splatt
