// Chains-on-chains partitioning (CCP): optimal contiguous partition of
// weighted items onto p workers minimizing the max part weight.
// Capability parity: reference src/thread_partition.c (lprobe:83-121,
// bisection:124-151, partition_weighted:156-195, partition_simple:198-217).
// Fresh design: integer binary search over the bottleneck bound (exact,
// no epsilon), probe via upper_bound on the prefix sums.
#pragma once

#include "types.hpp"

namespace splatt {

// returns true iff items can be split into <= nparts contiguous parts each
// of weight <= bound; fills parts[0..nparts] boundaries greedily.
bool ccp_probe(const int64_t * prefix, int64_t n, int nparts, int64_t bound,
               int64_t * parts);

// optimal boundaries (size nparts+1, parts[0]=0, parts[nparts]=n)
std::vector<int64_t> partition_weighted(const int64_t * weights, int64_t n,
                                        int nparts, int64_t * bottleneck = nullptr);

std::vector<int64_t> partition_simple(int64_t n, int nparts);

}  // namespace splatt
