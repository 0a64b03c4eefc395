#include "cpd.hpp"
#include <cstring>
#include <cmath>

namespace splatt {

namespace {
// counter-based splitmix64 hash -> uniform [0,1); partition invariant
inline uint64_t splitmix64(uint64_t x) {
  x += 0x9E3779B97F4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}
}  // namespace

template <typename V>
void seeded_factor_init(V * A, idx_t nrows, int rank, idx_t row0,
                        uint64_t seed, int mode) {
  #pragma omp parallel for schedule(static)
  for (int64_t i = 0; i < (int64_t)nrows; ++i) {
    const uint64_t gi = row0 + (uint64_t)i;
    for (int f = 0; f < rank; ++f) {
      const uint64_t h = splitmix64(seed ^ (0x51ED2701ull * (uint64_t)(mode + 1))
                                    ^ (gi * 0x100000001B3ull) ^ (uint64_t)f);
      A[(idx_t)i * rank + f] = (V)((double)(h >> 11) * 0x1.0p-53);
    }
  }
}

template <typename V>
Kruskal<V> cpd_als(const CsfSet<V> & set, int rank, const Options & opts,
                   std::function<void(int, double, double, double)> iter_cb) {
  const Csf<V> & c0 = set.csfs[0];
  const int nm = c0.nmodes;
  const int F = rank;

  Kruskal<V> k;
  k.nmodes = nm;
  k.rank = F;
  for (int m = 0; m < nm; ++m) k.dims[m] = c0.dims[m];
  k.lambda.assign(F, (V)1);

  idx_t maxdim = 0;
  for (int m = 0; m < nm; ++m) {
    k.factors[m].resize(c0.dims[m] * (idx_t)F);
    seeded_factor_init(k.factors[m].data(), c0.dims[m], F, 0,
                       opts.seed ? opts.seed : 0x5eed5eedull, m);
    maxdim = std::max(maxdim, c0.dims[m]);
  }

  std::array<std::vector<V>, MAX_NMODES> grams;
  std::array<const V*, MAX_NMODES> gram_ptrs{};
  std::array<const V*, MAX_NMODES> mat_ptrs{};
  for (int m = 0; m < nm; ++m) {
    grams[m].resize((size_t)F * F);
    mat_ata(k.factors[m].data(), c0.dims[m], F, grams[m].data());
    gram_ptrs[m] = grams[m].data();
    mat_ptrs[m] = k.factors[m].data();
  }

  const double normX = csf_frobsq(c0);
  std::vector<V> mttkrp_buf(maxdim * (idx_t)F);
  std::vector<V> G((size_t)F * F);
  double fit = 0, oldfit = 0;

  for (idx_t it = 0; it < opts.max_iters; ++it) {
    for (int m = 0; m < nm; ++m) {
      const Csf<V> & c = set.csfs[set.mode_csf[m]];
      mttkrp_csf_cpu(c, mat_ptrs.data(), mttkrp_buf.data(), m, F, opts.nthreads);
      std::memcpy(k.factors[m].data(), mttkrp_buf.data(),
                  sizeof(V) * c0.dims[m] * F);
      gram_hadamard(gram_ptrs.data(), nm, m, F, G.data());
      solve_normals(k.factors[m].data(), c0.dims[m], F, G.data(),
                    (V)opts.regularize);
      // 2-norm on first iteration, max-norm after (reference cpd.c:343-347)
      mat_normalize(k.factors[m].data(), c0.dims[m], F, k.lambda.data(),
                    it == 0 ? 0 : 1);
      mat_ata(k.factors[m].data(), c0.dims[m], F, grams[m].data());
    }

    // fit from the last mode's pre-solve MTTKRP output:
    // <X,K> = sum_f lambda_f * sum_i buf[i,f] * A_last[i,f]
    const int lastm = nm - 1;
    double inner = 0;
    {
      const V * A = k.factors[lastm].data();
      const idx_t n = c0.dims[lastm];
      #pragma omp parallel for schedule(static) reduction(+:inner)
      for (int64_t i = 0; i < (int64_t)n; ++i)
        for (int f = 0; f < F; ++f)
          inner += (double)mttkrp_buf[(idx_t)i * F + f] * (double)A[(idx_t)i * F + f]
                   * (double)k.lambda[f];
    }
    // ||K||^2 = lambda^T (hadamard of all grams) lambda
    gram_hadamard(gram_ptrs.data(), nm, -1, F, G.data());
    double knorm = 0;
    for (int a = 0; a < F; ++a)
      for (int b = 0; b < F; ++b)
        knorm += (double)G[(size_t)a * F + b] * (double)k.lambda[a] * (double)k.lambda[b];

    const double residual = std::sqrt(std::max(0.0, normX + knorm - 2 * inner));
    fit = 1.0 - residual / std::sqrt(normX);
    k.niters = (int)it + 1;
    if (iter_cb) iter_cb((int)it, fit, fit - oldfit, residual);
    if (it > 0 && std::abs(fit - oldfit) < opts.tolerance) break;
    oldfit = fit;
  }
  k.fit = fit;
  // post-process: unit 2-norm columns, scales folded into lambda
  // (reference cpd_post_process, cpd.c:391-411)
  for (int m = 0; m < nm; ++m) {
    std::vector<V> norms(F);
    mat_normalize(k.factors[m].data(), c0.dims[m], F, norms.data(), 0);
    for (int f = 0; f < F; ++f) k.lambda[f] *= norms[f];
  }
  return k;
}

template void seeded_factor_init<float>(float*, idx_t, int, idx_t, uint64_t, int);
template void seeded_factor_init<double>(double*, idx_t, int, idx_t, uint64_t, int);
template Kruskal<float> cpd_als<float>(const CsfSet<float>&, int, const Options&,
                                       std::function<void(int,double,double,double)>);
template Kruskal<double> cpd_als<double>(const CsfSet<double>&, int, const Options&,
                                         std::function<void(int,double,double,double)>);

}  // namespace splatt
