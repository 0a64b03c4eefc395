// C API implementation over the C++17 core (torch-free).
// Parity target: reference src/splatt_lapack.h-free API layer
// (lib/CMakeLists.txt builds libsplatt; api entry points cited in
// csrc/capi/splatt.h). The opaque splatt_csf handle wraps the core's
// CsfSet so the 1/2/nmodes allocation policies behave like the
// reference's csf_alloc (csf.c:770-814).
#include "splatt.h"

#include <cstring>
#include <new>
#include <string>

#include "../core/types.hpp"
#include "../core/sptensor.hpp"
#include "../core/csf.hpp"
#include "../core/mttkrp_cpu.hpp"
#include "../core/cpd.hpp"
#include "../core/io.hpp"

using namespace splatt;

namespace splatt {
// device engine (csrc/capi/capi_gpu.cpp, linked with the HIP kernels)
bool capi_gpu_available();
template <typename V>
Kruskal<V> cpd_als_gpu(const CsfSet<V> &, int, const Options &);
template <typename V>
int mttkrp_gpu(const CsfSet<V> &, void **, int, int, const V * const *,
               V *);
void free_dev_csf_cache(void *);
}

struct splatt_csf {
  CsfSet<double> set;
  void * dev_cache = nullptr;   // device-resident streams (capi_gpu.cpp)
};

static Options opts_from_array(const double * o) {
  Options opt;
  if (!o) return opt;
  opt.tolerance = o[SPLATT_OPTION_TOLERANCE];
  opt.regularize = o[SPLATT_OPTION_REGULARIZE];
  opt.max_iters = (idx_t)o[SPLATT_OPTION_NITER];
  opt.verbosity = (int)o[SPLATT_OPTION_VERBOSITY];
  opt.nthreads = (int)o[SPLATT_OPTION_NTHREADS];
  opt.seed = (uint64_t)o[SPLATT_OPTION_RANDSEED];
  switch ((int)o[SPLATT_OPTION_CSF_ALLOC]) {
    case SPLATT_CSF_ONEMODE: opt.csf_alloc = CsfAlloc::ONEMODE; break;
    case SPLATT_CSF_ALLMODE: opt.csf_alloc = CsfAlloc::ALLMODE; break;
    default: opt.csf_alloc = CsfAlloc::TWOMODE; break;
  }
  return opt;
}

extern "C" {

double * splatt_default_opts(void) {
  double * o = (double*)aligned_alloc64(sizeof(double) * SPLATT_OPTION_NOPTIONS);
  o[SPLATT_OPTION_TOLERANCE] = 1e-5;
  o[SPLATT_OPTION_NITER] = 50;
  o[SPLATT_OPTION_VERBOSITY] = SPLATT_VERBOSITY_LOW;
  o[SPLATT_OPTION_NTHREADS] = 0;
  o[SPLATT_OPTION_RANDSEED] = (double)0x5EED5EEDull;
  o[SPLATT_OPTION_CSF_ALLOC] = SPLATT_CSF_TWOMODE;
  o[SPLATT_OPTION_REGULARIZE] = 0.0;
  o[SPLATT_OPTION_TILE] = 0.0;
  o[SPLATT_OPTION_TILELEVEL] = 1.0;
  o[SPLATT_OPTION_PRIVTHRESH] = 0.02;
  o[SPLATT_OPTION_DECOMP] = SPLATT_DECOMP_MEDIUM;
  o[SPLATT_OPTION_COMM] = SPLATT_COMM_ALL2ALL;
  return o;
}

void splatt_free_opts(double * opts) { aligned_free64(opts); }

int splatt_csf_load(const char * fname, splatt_idx_t * nmodes,
                    splatt_csf ** tensors, const double * options) {
  try {
    auto opt = opts_from_array(options);
    auto tt = tensor_load<double>(fname);
    *nmodes = (splatt_idx_t)tt.nmodes;
    auto * h = new splatt_csf{csf_alloc(tt, opt)};
    *tensors = h;
    return SPLATT_SUCCESS;
  } catch (const std::bad_alloc &) {
    return SPLATT_ERROR_NOMEMORY;
  } catch (...) {
    return SPLATT_ERROR_BADINPUT;
  }
}

int splatt_csf_convert(splatt_idx_t nmodes, splatt_idx_t nnz,
                       splatt_idx_t ** inds, splatt_val_t * vals,
                       splatt_csf ** tensors, const double * options) {
  try {
    auto opt = opts_from_array(options);
    idx_t dims[MAX_NMODES] = {0};
    for (splatt_idx_t m = 0; m < nmodes; ++m)
      for (splatt_idx_t i = 0; i < nnz; ++i)
        dims[m] = std::max(dims[m], inds[m][i] + 1);
    SpTensor<double> tt((int)nmodes, nnz, dims);
    for (splatt_idx_t m = 0; m < nmodes; ++m)
      std::memcpy(tt.ind[m].data(), inds[m], sizeof(idx_t) * nnz);
    std::memcpy(tt.vals.data(), vals, sizeof(double) * nnz);
    *tensors = new splatt_csf{csf_alloc(tt, opt)};
    return SPLATT_SUCCESS;
  } catch (const std::bad_alloc &) {
    return SPLATT_ERROR_NOMEMORY;
  } catch (...) {
    return SPLATT_ERROR_BADINPUT;
  }
}

void splatt_free_csf(splatt_csf * tensors, const double *) {
  if (tensors && tensors->dev_cache)
    free_dev_csf_cache(tensors->dev_cache);
  delete tensors;
}

splatt_idx_t splatt_csf_nmodes(const splatt_csf * csf) {
  return (splatt_idx_t)csf->set.csfs[0].nmodes;
}
splatt_idx_t splatt_csf_nnz(const splatt_csf * csf) {
  return (splatt_idx_t)csf->set.csfs[0].nnz;
}
void splatt_csf_dims(const splatt_csf * csf, splatt_idx_t * dims) {
  const auto & c = csf->set.csfs[0];
  for (int m = 0; m < c.nmodes; ++m) dims[m] = c.dims[m];
}

int splatt_cpd_als(const splatt_csf * tensors, splatt_idx_t nfactors,
                   const double * options, splatt_kruskal * factored) {
  try {
    auto opt = opts_from_array(options);
    // HIP engine when an MI355X is visible (SPLATT_CAPI_CPU=1 forces the
    // host core); errors surface loudly rather than silently falling back
    auto k = capi_gpu_available()
                 ? cpd_als_gpu<double>(tensors->set, (int)nfactors, opt)
                 : cpd_als(tensors->set, (int)nfactors, opt);
    factored->rank = nfactors;
    factored->nmodes = (splatt_idx_t)k.nmodes;
    factored->fit = k.fit;
    factored->lambda =
        (splatt_val_t*)aligned_alloc64(sizeof(double) * nfactors);
    std::memcpy(factored->lambda, k.lambda.data(), sizeof(double) * nfactors);
    for (int m = 0; m < k.nmodes; ++m) {
      factored->dims[m] = k.dims[m];
      const size_t bytes = sizeof(double) * k.dims[m] * nfactors;
      factored->factors[m] = (splatt_val_t*)aligned_alloc64(bytes);
      std::memcpy(factored->factors[m], k.factors[m].data(), bytes);
    }
    return SPLATT_SUCCESS;
  } catch (const std::bad_alloc &) {
    return SPLATT_ERROR_NOMEMORY;
  } catch (...) {
    return SPLATT_ERROR_BADINPUT;
  }
}

void splatt_free_kruskal(splatt_kruskal * factored) {
  aligned_free64(factored->lambda);
  for (splatt_idx_t m = 0; m < factored->nmodes; ++m)
    aligned_free64(factored->factors[m]);
}

int splatt_mttkrp(splatt_idx_t mode, splatt_idx_t ncolumns,
                  const splatt_csf * tensors, splatt_val_t ** matrices,
                  splatt_val_t * matout, const double * options) {
  try {
    (void)options;
    const auto & set = tensors->set;
    // HIP engine when a GPU is visible (device CSF streams cached on the
    // handle after the first call); CPU core otherwise or on fallback
    if (capi_gpu_available()) {
      auto * h = const_cast<splatt_csf *>(tensors);
      if (mttkrp_gpu<double>(set, &h->dev_cache, (int)mode, (int)ncolumns,
                             (double const * const *)matrices,
                             matout) == 0)
        return SPLATT_SUCCESS;
    }
    const auto & c = set.csfs[set.mode_csf[mode]];
    mttkrp_csf_cpu(c, (double const * const *)matrices, matout, (int)mode,
                   (int)ncolumns);
    return SPLATT_SUCCESS;
  } catch (...) {
    return SPLATT_ERROR_BADINPUT;
  }
}

splatt_mttkrp_ws * splatt_mttkrp_alloc_ws(const splatt_csf * tensors,
                                          splatt_idx_t, const double *) {
  auto * ws = (splatt_mttkrp_ws*)aligned_alloc64(sizeof(splatt_mttkrp_ws));
  ws->num_csf = (splatt_idx_t)tensors->set.csfs.size();
  for (int m = 0; m < tensors->set.csfs[0].nmodes; ++m)
    ws->mode_csf_map[m] = (splatt_idx_t)tensors->set.mode_csf[m];
  return ws;
}
void splatt_mttkrp_free_ws(splatt_mttkrp_ws * ws) { aligned_free64(ws); }

int splatt_gpu_available(void) {
  return splatt::capi_gpu_available() ? 1 : 0;
}

int splatt_version_major(void) { return 0; }
int splatt_version_minor(void) { return 1; }
int splatt_version_subminor(void) { return 0; }

}  // extern "C"
