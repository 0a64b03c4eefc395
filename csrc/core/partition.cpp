#include "partition.hpp"
#include <algorithm>
#include <numeric>

namespace splatt {

bool ccp_probe(const int64_t * prefix, int64_t n, int nparts, int64_t bound,
               int64_t * parts) {
  parts[0] = 0;
  int64_t pos = 0;
  for (int p = 1; p <= nparts; ++p) {
    // furthest boundary with part weight <= bound
    const int64_t target = prefix[pos] + bound;
    pos = std::upper_bound(prefix + pos, prefix + n + 1, target)
          - prefix - 1;
    parts[p] = pos;
    if (pos == n) {
      for (int q = p + 1; q <= nparts; ++q) parts[q] = n;
      return true;
    }
  }
  return parts[nparts] == n;
}

std::vector<int64_t> partition_weighted(const int64_t * weights, int64_t n,
                                        int nparts, int64_t * bottleneck) {
  std::vector<int64_t> prefix(n + 1, 0);
  int64_t maxw = 0;
  for (int64_t i = 0; i < n; ++i) {
    prefix[i + 1] = prefix[i] + weights[i];
    maxw = std::max(maxw, weights[i]);
  }
  const int64_t total = prefix[n];
  std::vector<int64_t> parts(nparts + 1);
  int64_t lo = std::max(maxw, (total + nparts - 1) / std::max(nparts, 1));
  int64_t hi = total;
  while (lo < hi) {  // smallest feasible bottleneck
    const int64_t mid = lo + (hi - lo) / 2;
    if (ccp_probe(prefix.data(), n, nparts, mid, parts.data())) hi = mid;
    else lo = mid + 1;
  }
  ccp_probe(prefix.data(), n, nparts, lo, parts.data());
  if (bottleneck) *bottleneck = lo;
  return parts;
}

std::vector<int64_t> partition_simple(int64_t n, int nparts) {
  std::vector<int64_t> parts(nparts + 1);
  for (int p = 0; p <= nparts; ++p)
    parts[p] = p * (n / nparts) + std::min<int64_t>(p, n % nparts);
  parts[nparts] = n;
  return parts;
}

}  // namespace splatt
