// Host MTTKRP kernels: COO streaming oracle + CSF tree walkers at any
// output depth. Capability parity: reference src/mttkrp.c
// (mttkrp_stream:1697-1757 = the gold oracle; root/intl/leaf families
// mttkrp.c:390-1278). Fresh design: one recursive walker per output class
// instead of six hand-specialized loop nests; OpenMP over root nodes.
#pragma once

#include "csf.hpp"

namespace splatt {

// mats[m]: row-major dims[m] x rank. out: row-major dims[mode] x rank, zeroed
// inside. COO streaming gold oracle (serial, deterministic).
template <typename V>
void mttkrp_stream(const SpTensor<V> & tt, V const * const * mats,
                   V * out, int mode, int rank);

// CSF MTTKRP with output at mode `mode` (dispatches on depth in `c`).
template <typename V>
void mttkrp_csf_cpu(const Csf<V> & c, V const * const * mats,
                    V * out, int mode, int rank, int nthreads = 0);

}  // namespace splatt
