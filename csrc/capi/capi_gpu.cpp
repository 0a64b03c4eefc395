// Torch-free device CPD-ALS for the C API (splatt_cpd_als): when an
// MI355X is visible, the flagship HIP engine — the flat CDNA4 MTTKRP
// kernels (csrc/hip/mttkrp_flat.hip), the one-workgroup SPD inverse and
// LDS rowsolve GEMM (csrc/hip/dense_kernels.hip) — runs under the same
// splatt_* C surface the reference exposes (reference
// include/splatt/api_factorization.h:41-46). Host orchestration only;
// all O(nnz) and O(dims*F) work stays on the device.
//
// Math mirrors csrc/core/cpd.cpp (same seeded init, normalize schedule,
// fit formula), so fits agree with both the CPU C path and the Python
// device driver up to atomic summation order.
#include <hip/hip_runtime.h>

#include <cmath>
#include <cstring>
#include <stdexcept>
#include <vector>

#include "../core/types.hpp"
#include "../core/csf.hpp"
#include "../core/cpd.hpp"
#include "../core/matrix.hpp"

extern "C" {
void splatt_hip_mttkrp_flat_f64(
    const int32_t * key, const int32_t * const * idx,
    const double * const * mats, const double * vals, int64_t nnz,
    double * out, int rank, int nother, void * stream);
int splatt_hip_rowsolve_f64(const double *, const double *, double *,
                            int64_t, int, void *);
void splatt_hip_gram_f64(const double *, int64_t, int, double *, void *);
void splatt_hip_spd_inverse_f64(const double *, double *, int, void *);
void splatt_hip_colacc_f64(const double *, int64_t, int, int, double *,
                           void *);
void splatt_hip_colscale_f64(double *, int64_t, int, const double *, void *);
void splatt_hip_coldot_f64(const double *, const double *, int64_t, int,
                           double *, void *);
}

namespace splatt {

namespace {

void hip_check(hipError_t e, const char * what) {
  if (e != hipSuccess)
    throw std::runtime_error(std::string(what) + ": " +
                             hipGetErrorString(e));
}

template <typename T>
T * dupload(const std::vector<T> & h) {
  T * d = nullptr;
  hip_check(hipMalloc(&d, sizeof(T) * std::max<size_t>(1, h.size())),
            "hipMalloc");
  hip_check(hipMemcpy(d, h.data(), sizeof(T) * h.size(),
                      hipMemcpyHostToDevice), "hipMemcpy H2D");
  return d;
}

// per-nonzero ancestor label expansion at `level` (the flat kernels'
// index streams; host analog of splatt_amd/csf.py ancestor_expand)
template <typename V>
std::vector<int32_t> expand_level(const Csf<V> & c, int level) {
  const int nm = c.nmodes;
  std::vector<int32_t> lab(c.nnz);
  if (level == nm - 1) {
    #pragma omp parallel for schedule(static)
    for (int64_t i = 0; i < (int64_t)c.nnz; ++i)
      lab[i] = (int32_t)c.fids[nm - 1][i];
    return lab;
  }
  const idx_t nf = c.nfibs[level];
  const bool ident = c.fids[level].empty();
  #pragma omp parallel for schedule(static)
  for (int64_t k = 0; k < (int64_t)nf; ++k) {
    // nnz span of node k: chain fptr down to the leaf level
    int64_t s = k, e = k + 1;
    for (int j = level; j < nm - 1; ++j) {
      s = c.fptr[j][s];
      e = c.fptr[j][e];
    }
    const int32_t v = ident ? (int32_t)k : (int32_t)c.fids[level][k];
    for (int64_t p = s; p < e; ++p) lab[p] = v;
  }
  return lab;
}

struct DevBuf {
  double * p = nullptr;
  ~DevBuf() { if (p) (void)hipFree(p); }
  void alloc(size_t elems) {
    hip_check(hipMalloc(&p, sizeof(double) * std::max<size_t>(1, elems)),
              "hipMalloc");
  }
};

struct DevIBuf {
  int32_t * p = nullptr;
  ~DevIBuf() { if (p) (void)hipFree(p); }
};

}  // namespace

bool capi_gpu_available() {
  if (const char * e = getenv("SPLATT_CAPI_CPU"))
    if (e[0] == '1') return false;
  int n = 0;
  return hipGetDeviceCount(&n) == hipSuccess && n > 0;
}

// ---- device-resident CSF cache for repeated splatt_mttkrp calls:
// label streams + values uploaded once per handle, factors per call
struct DevCsfCache {
  int ncsf = 0;
  int nmodes = 0;
  std::vector<std::vector<DevIBuf>> labs;   // [csf][level]
  std::vector<DevBuf> vals;                 // [csf]
};

void free_dev_csf_cache(void * p) {
  delete reinterpret_cast<DevCsfCache *>(p);
}

template <typename V>
int mttkrp_gpu(const CsfSet<V> & set, void ** cache_slot, int mode,
               int rank, const V * const * mats_host, V * out_host);

template <>
int mttkrp_gpu<double>(const CsfSet<double> & set, void ** cache_slot,
                       int mode, int rank,
                       const double * const * mats_host,
                       double * out_host) {
  const Csf<double> & c0 = set.csfs[0];
  const int nm = c0.nmodes;
  if (rank > 64) return -1;
  for (int m = 0; m < nm; ++m)
    if (c0.dims[m] > 0x7FFFFFFFull) return -1;
  hipStream_t st = nullptr;
  auto * cache = reinterpret_cast<DevCsfCache *>(*cache_slot);
  if (!cache) {
    cache = new DevCsfCache;
    cache->ncsf = (int)set.csfs.size();
    cache->nmodes = nm;
    cache->labs.resize(cache->ncsf);
    cache->vals.resize(cache->ncsf);
    for (int ci = 0; ci < cache->ncsf; ++ci) {
      const auto & c = set.csfs[ci];
      cache->labs[ci].resize(nm);
      for (int l = 0; l < nm; ++l) {
        auto h = expand_level(c, l);
        cache->labs[ci][l].p = (int32_t*)dupload(h);
      }
      cache->vals[ci].alloc(c.nnz);
      hip_check(hipMemcpy(cache->vals[ci].p, c.vals.data(),
                          sizeof(double) * c.nnz, hipMemcpyHostToDevice),
                "vals H2D");
    }
    *cache_slot = cache;
  }
  const int ci = set.mode_csf[mode];
  const auto & c = set.csfs[ci];
  const int depth = set.mode_depth[mode];
  std::vector<DevBuf> dmats(nm);
  for (int m = 0; m < nm; ++m) {
    dmats[m].alloc(c0.dims[m] * (size_t)rank);
    hip_check(hipMemcpyAsync(dmats[m].p, mats_host[m],
                             sizeof(double) * c0.dims[m] * rank,
                             hipMemcpyHostToDevice, st), "mats H2D");
  }
  DevBuf dout;
  dout.alloc(c0.dims[mode] * (size_t)rank);
  hip_check(hipMemsetAsync(dout.p, 0,
                           sizeof(double) * c0.dims[mode] * rank, st),
            "memset");
  const int32_t * idxp[8] = {nullptr};
  const double * matp[8] = {nullptr};
  int t = 0;
  for (int l = 0; l < nm; ++l) {
    if (l == depth) continue;
    idxp[t] = cache->labs[ci][l].p;
    matp[t] = dmats[c.dim_perm[l]].p;
    ++t;
  }
  splatt_hip_mttkrp_flat_f64(cache->labs[ci][depth].p, idxp, matp,
                             cache->vals[ci].p, (int64_t)c.nnz, dout.p,
                             rank, nm - 1, st);
  hip_check(hipStreamSynchronize(st), "sync");
  hip_check(hipMemcpy(out_host, dout.p,
                      sizeof(double) * c0.dims[mode] * rank,
                      hipMemcpyDeviceToHost), "out D2H");
  return 0;
}

template <typename V>
Kruskal<V> cpd_als_gpu(const CsfSet<V> & set, int rank, const Options & opts);

template <>
Kruskal<double> cpd_als_gpu<double>(const CsfSet<double> & set, int rank,
                                    const Options & opts) {
  const Csf<double> & c0 = set.csfs[0];
  const int nm = c0.nmodes;
  const int F = rank;
  if (F > 64)
    throw std::runtime_error("device CPD supports rank <= 64");
  for (int m = 0; m < nm; ++m)
    if (c0.dims[m] > 0x7FFFFFFFull)
      throw std::runtime_error("device CPD: mode dims must fit int32");
  const bool spec = (F == 4 || F == 8 || F == 16 || F == 32 || F == 64);
  hipStream_t st = nullptr;

  // upload every distinct CSF: per-level label streams + values
  const int ncsf = (int)set.csfs.size();
  std::vector<std::vector<DevIBuf>> labs(ncsf);
  std::vector<DevBuf> dvals(ncsf);
  for (int ci = 0; ci < ncsf; ++ci) {
    const auto & c = set.csfs[ci];
    labs[ci].resize(nm);
    for (int l = 0; l < nm; ++l) {
      auto h = expand_level(c, l);
      labs[ci][l].p = (int32_t*)dupload(h);
    }
    dvals[ci].alloc(c.nnz);
    hip_check(hipMemcpy(dvals[ci].p, c.vals.data(),
                        sizeof(double) * c.nnz, hipMemcpyHostToDevice),
              "vals H2D");
  }

  Kruskal<double> k;
  k.nmodes = nm;
  k.rank = F;
  k.lambda.assign(F, 1.0);
  idx_t maxdim = 0;
  std::vector<DevBuf> dA(nm);
  std::array<std::vector<double>, MAX_NMODES> grams;
  for (int m = 0; m < nm; ++m) {
    k.dims[m] = c0.dims[m];
    k.factors[m].resize(c0.dims[m] * (idx_t)F);
    seeded_factor_init(k.factors[m].data(), c0.dims[m], F, 0,
                       opts.seed ? opts.seed : 0x5eed5eedull, m);
    dA[m].alloc(c0.dims[m] * (size_t)F);
    hip_check(hipMemcpy(dA[m].p, k.factors[m].data(),
                        sizeof(double) * c0.dims[m] * F,
                        hipMemcpyHostToDevice), "factor H2D");
    grams[m].resize((size_t)F * F);
    mat_ata(k.factors[m].data(), c0.dims[m], F, grams[m].data());
    maxdim = std::max(maxdim, c0.dims[m]);
  }

  const double normX = csf_frobsq(c0);
  DevBuf dbuf, dG, dGinv, dsmall;
  dbuf.alloc(maxdim * (size_t)F);
  dG.alloc((size_t)F * F);
  dGinv.alloc((size_t)F * F);
  dsmall.alloc(F);
  std::vector<double> hbuf;           // host staging for non-spec solve
  std::vector<double> G((size_t)F * F), lamh(F);

  double fit = 0, oldfit = 0;
  for (idx_t it = 0; it < opts.max_iters; ++it) {
    for (int m = 0; m < nm; ++m) {
      const int ci = set.mode_csf[m];
      const auto & c = set.csfs[ci];
      const int depth = set.mode_depth[m];
      const idx_t n = c0.dims[m];
      // flat MTTKRP at the dispatch depth (any key order: one atomic
      // per output-key run)
      hip_check(hipMemsetAsync(dbuf.p, 0, sizeof(double) * n * F, st),
                "memset");
      const int32_t * idxp[8] = {nullptr};
      const double * matp[8] = {nullptr};
      int t = 0;
      for (int l = 0; l < nm; ++l) {
        if (l == depth) continue;
        idxp[t] = labs[ci][l].p;
        matp[t] = dA[c.dim_perm[l]].p;
        ++t;
      }
      splatt_hip_mttkrp_flat_f64(labs[ci][depth].p, idxp, matp, dvals[ci].p,
                                 (int64_t)c.nnz, dbuf.p, F, nm - 1, st);
      // normal equations: G = hadamard of other grams (+reg I)
      std::array<const double*, MAX_NMODES> gp{};
      for (int o = 0; o < nm; ++o) gp[o] = grams[o].data();
      gram_hadamard(gp.data(), nm, m, F, G.data());
      if (opts.regularize != 0)
        for (int f = 0; f < F; ++f)
          G[(size_t)f * F + f] += (double)opts.regularize;
      if (spec) {
        hip_check(hipMemcpyAsync(dG.p, G.data(), sizeof(double) * F * F,
                                 hipMemcpyHostToDevice, st), "G H2D");
        splatt_hip_spd_inverse_f64(dG.p, dGinv.p, F, st);
        if (splatt_hip_rowsolve_f64(dbuf.p, dGinv.p, dA[m].p, (int64_t)n,
                                    F, st) != 0)
          throw std::runtime_error("rowsolve failed");
      } else {
        // generic rank: Cholesky solve on host (small F, n-row RHS)
        hbuf.resize(n * (size_t)F);
        hip_check(hipStreamSynchronize(st), "sync");
        hip_check(hipMemcpy(hbuf.data(), dbuf.p, sizeof(double) * n * F,
                            hipMemcpyDeviceToHost), "buf D2H");
        std::vector<double> sol = hbuf;
        solve_normals(sol.data(), n, F, G.data(), 0.0);
        hip_check(hipMemcpy(dA[m].p, sol.data(), sizeof(double) * n * F,
                            hipMemcpyHostToDevice), "sol H2D");
      }
      // normalize: 2-norm on it 0, max-norm after (reference
      // cpd.c:343-347); lambda math on host (length F)
      hip_check(hipMemsetAsync(dsmall.p, 0, sizeof(double) * F, st),
                "memset");
      splatt_hip_colacc_f64(dA[m].p, (int64_t)n, F, it == 0 ? 0 : 1,
                            dsmall.p, st);
      hip_check(hipStreamSynchronize(st), "sync");
      hip_check(hipMemcpy(lamh.data(), dsmall.p, sizeof(double) * F,
                          hipMemcpyDeviceToHost), "lam D2H");
      for (int f = 0; f < F; ++f) {
        double lv = it == 0 ? std::sqrt(lamh[f])
                            : std::max(lamh[f], 1.0);
        lamh[f] = (lv == 0.0) ? 1.0 : lv;
        k.lambda[f] = lamh[f];
      }
      hip_check(hipMemcpyAsync(dsmall.p, lamh.data(), sizeof(double) * F,
                               hipMemcpyHostToDevice, st), "lam H2D");
      splatt_hip_colscale_f64(dA[m].p, (int64_t)n, F, dsmall.p, st);
      // fresh Gram of the updated factor
      hip_check(hipMemsetAsync(dG.p, 0, sizeof(double) * F * F, st),
                "memset");
      splatt_hip_gram_f64(dA[m].p, (int64_t)n, F, dG.p, st);
      hip_check(hipStreamSynchronize(st), "sync");
      hip_check(hipMemcpy(grams[m].data(), dG.p, sizeof(double) * F * F,
                          hipMemcpyDeviceToHost), "gram D2H");
    }

    // fit: inner product from the last mode's pre-solve MTTKRP output
    const int lastm = nm - 1;
    hip_check(hipMemsetAsync(dsmall.p, 0, sizeof(double) * F, st), "memset");
    splatt_hip_coldot_f64(dbuf.p, dA[lastm].p, (int64_t)c0.dims[lastm], F,
                          dsmall.p, st);
    hip_check(hipStreamSynchronize(st), "sync");
    hip_check(hipMemcpy(lamh.data(), dsmall.p, sizeof(double) * F,
                        hipMemcpyDeviceToHost), "inner D2H");
    double inner = 0;
    for (int f = 0; f < F; ++f) inner += lamh[f] * k.lambda[f];
    std::array<const double*, MAX_NMODES> gp{};
    for (int o = 0; o < nm; ++o) gp[o] = grams[o].data();
    gram_hadamard(gp.data(), nm, -1, F, G.data());
    double knorm = 0;
    for (int a = 0; a < F; ++a)
      for (int b = 0; b < F; ++b)
        knorm += G[(size_t)a * F + b] * k.lambda[a] * k.lambda[b];
    const double residual = std::sqrt(std::max(0.0, normX + knorm
                                               - 2 * inner));
    fit = 1.0 - residual / std::sqrt(normX);
    k.niters = (int)it + 1;
    if (it > 0 && std::abs(fit - oldfit) < opts.tolerance) break;
    oldfit = fit;
  }
  k.fit = fit;

  // post-process: unit 2-norm columns, scales folded into lambda
  for (int m = 0; m < nm; ++m) {
    const idx_t n = c0.dims[m];
    hip_check(hipMemsetAsync(dsmall.p, 0, sizeof(double) * F, st), "memset");
    splatt_hip_colacc_f64(dA[m].p, (int64_t)n, F, 0, dsmall.p, st);
    hip_check(hipStreamSynchronize(st), "sync");
    hip_check(hipMemcpy(lamh.data(), dsmall.p, sizeof(double) * F,
                        hipMemcpyDeviceToHost), "norm D2H");
    for (int f = 0; f < F; ++f) {
      double nv = std::sqrt(lamh[f]);
      lamh[f] = (nv == 0.0) ? 1.0 : nv;
      k.lambda[f] *= lamh[f];
    }
    hip_check(hipMemcpyAsync(dsmall.p, lamh.data(), sizeof(double) * F,
                             hipMemcpyHostToDevice, st), "norm H2D");
    splatt_hip_colscale_f64(dA[m].p, (int64_t)n, F, dsmall.p, st);
    hip_check(hipStreamSynchronize(st), "sync");
    hip_check(hipMemcpy(k.factors[m].data(), dA[m].p,
                        sizeof(double) * n * F, hipMemcpyDeviceToHost),
              "factor D2H");
  }
  return k;
}

}  // namespace splatt
