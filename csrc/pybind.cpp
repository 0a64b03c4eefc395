// Python boundary of the MI355X-native sparse tensor engine.
// CPU core functions exposed over torch tensors; HIP kernel launchers are
// declared extern "C" (defined in csrc/hip/*.hip, compiled by hipcc for
// gfx950 and linked in) and take raw device pointers + the current stream,
// so this translation unit needs no HIP headers.
#include <torch/extension.h>
#include <vector>
#include <string>

#include "core/types.hpp"
#include "core/sptensor.hpp"
#include "core/csf.hpp"
#include "core/mttkrp_cpu.hpp"
#include "core/matrix.hpp"
#include "core/cpd.hpp"
#include "core/io.hpp"
#include "core/partition.hpp"

namespace sp = splatt;
using torch::Tensor;

// ---------------------------------------------------------------- helpers

static sp::Options opts_from_dict(const py::dict & d) {
  sp::Options o;
  if (d.contains("tolerance")) o.tolerance = d["tolerance"].cast<double>();
  if (d.contains("max_iters")) o.max_iters = d["max_iters"].cast<uint64_t>();
  if (d.contains("seed")) o.seed = d["seed"].cast<uint64_t>();
  if (d.contains("nthreads")) o.nthreads = d["nthreads"].cast<int>();
  if (d.contains("regularize"))
    o.regularize = d["regularize"].cast<double>();
  if (d.contains("csf_alloc")) {
    const std::string a = d["csf_alloc"].cast<std::string>();
    o.csf_alloc = a == "one" ? sp::CsfAlloc::ONEMODE
                : a == "all" ? sp::CsfAlloc::ALLMODE
                             : sp::CsfAlloc::TWOMODE;
  }
  return o;
}

template <typename V>
static sp::SpTensor<V> coo_from_torch(const Tensor & inds, const Tensor & vals,
                                      const std::vector<int64_t> & dims) {
  TORCH_CHECK(inds.dim() == 2, "inds must be [nmodes, nnz]");
  TORCH_CHECK(inds.scalar_type() == torch::kInt64, "inds must be int64");
  auto ic = inds.contiguous();
  auto vc = vals.contiguous();
  const int nm = (int)ic.size(0);
  const sp::idx_t nnz = (sp::idx_t)ic.size(1);
  sp::idx_t d[sp::MAX_NMODES];
  for (int m = 0; m < nm; ++m) d[m] = (sp::idx_t)dims[m];
  sp::SpTensor<V> tt(nm, nnz, d);
  const int64_t * ip = ic.data_ptr<int64_t>();
  const V * vp = vc.data_ptr<V>();
  for (int m = 0; m < nm; ++m)
    for (sp::idx_t i = 0; i < nnz; ++i) tt.ind[m][i] = (sp::idx_t)ip[m * nnz + i];
  std::copy(vp, vp + nnz, tt.vals.begin());
  return tt;
}

template <typename V>
static std::tuple<Tensor, Tensor, std::vector<int64_t>>
coo_to_torch(const sp::SpTensor<V> & tt) {
  const auto vopt = torch::TensorOptions().dtype(
      std::is_same<V, double>::value ? torch::kFloat64 : torch::kFloat32);
  Tensor inds = torch::empty({tt.nmodes, (int64_t)tt.nnz}, torch::kInt64);
  Tensor vals = torch::empty({(int64_t)tt.nnz}, vopt);
  int64_t * ip = inds.data_ptr<int64_t>();
  V * vp = vals.data_ptr<V>();
  for (int m = 0; m < tt.nmodes; ++m)
    for (sp::idx_t i = 0; i < tt.nnz; ++i) ip[m * tt.nnz + i] = (int64_t)tt.ind[m][i];
  std::copy(tt.vals.begin(), tt.vals.end(), vp);
  std::vector<int64_t> dims(tt.nmodes);
  for (int m = 0; m < tt.nmodes; ++m) dims[m] = (int64_t)tt.dims[m];
  return {inds, vals, dims};
}

// csf as a python dict of torch tensors
template <typename V>
static py::dict csf_to_py(const sp::Csf<V> & c) {
  py::dict d;
  const auto vopt = torch::TensorOptions().dtype(
      std::is_same<V, double>::value ? torch::kFloat64 : torch::kFloat32);
  py::list fptr, fids;
  for (int l = 0; l < c.nmodes; ++l) {
    if (l < c.nmodes - 1) {
      Tensor t = torch::empty({(int64_t)c.fptr[l].size()}, torch::kInt64);
      std::copy(c.fptr[l].begin(), c.fptr[l].end(), t.data_ptr<int64_t>());
      fptr.append(t);
    } else {
      fptr.append(py::none());
    }
    if (c.fids[l].empty()) {
      fids.append(py::none());
    } else {
      Tensor t = torch::empty({(int64_t)c.fids[l].size()}, torch::kInt32);
      const sp::fid_t * s = c.fids[l].data();
      int32_t * p = t.data_ptr<int32_t>();
      for (size_t i = 0; i < c.fids[l].size(); ++i) p[i] = (int32_t)s[i];
      fids.append(t);
    }
  }
  Tensor vals = torch::empty({(int64_t)c.vals.size()}, vopt);
  std::copy(c.vals.begin(), c.vals.end(), vals.data_ptr<V>());
  d["fptr"] = fptr;
  d["fids"] = fids;
  d["vals"] = vals;
  std::vector<int64_t> dims(c.nmodes), perm(c.nmodes), nfibs(c.nmodes);
  for (int l = 0; l < c.nmodes; ++l) {
    dims[l] = (int64_t)c.dims[l];
    perm[l] = c.dim_perm[l];
    nfibs[l] = (int64_t)c.nfibs[l];
  }
  d["dims"] = dims;
  d["dim_perm"] = perm;
  d["nfibs"] = nfibs;
  return d;
}

template <typename V>
static sp::Csf<V> csf_from_py(const py::dict & d) {
  sp::Csf<V> c;
  auto dims = d["dims"].cast<std::vector<int64_t>>();
  auto perm = d["dim_perm"].cast<std::vector<int64_t>>();
  c.nmodes = (int)dims.size();
  for (int l = 0; l < c.nmodes; ++l) {
    c.dims[l] = (sp::idx_t)dims[l];
    c.dim_perm[l] = (int)perm[l];
    c.dim_iperm[perm[l]] = l;
  }
  py::list fptr = d["fptr"], fids = d["fids"];
  for (int l = 0; l < c.nmodes; ++l) {
    if (!fptr[l].is_none()) {
      Tensor t = fptr[l].cast<Tensor>().contiguous();
      c.fptr[l].assign(t.data_ptr<int64_t>(), t.data_ptr<int64_t>() + t.numel());
    }
    if (!fids[l].is_none()) {
      Tensor t = fids[l].cast<Tensor>().contiguous();
      const int32_t * p = t.data_ptr<int32_t>();
      c.fids[l].resize(t.numel());
      for (int64_t i = 0; i < t.numel(); ++i) c.fids[l][i] = (sp::fid_t)p[i];
    }
  }
  Tensor vals = d["vals"].cast<Tensor>().contiguous();
  c.vals.assign(vals.data_ptr<V>(), vals.data_ptr<V>() + vals.numel());
  c.nnz = (sp::idx_t)c.vals.size();
  for (int l = 0; l < c.nmodes; ++l)
    c.nfibs[l] = l < c.nmodes - 1 ? (c.fptr[l].empty() ? 0 : c.fptr[l].size() - 1)
                                  : c.nnz;
  return c;
}

#define DTYPE_DISPATCH(vals_dtype, fn)                                     \
  ((vals_dtype) == torch::kFloat64 ? fn(double) : fn(float))

// ---------------------------------------------------------------- io

static py::object py_tensor_load(const std::string & path, const std::string & dtype) {
  if (dtype == "f32") {
    auto tt = sp::tensor_load<float>(path);
    auto [i, v, d] = coo_to_torch(tt);
    return py::make_tuple(i, v, d);
  }
  auto tt = sp::tensor_load<double>(path);
  auto [i, v, d] = coo_to_torch(tt);
  return py::make_tuple(i, v, d);
}

static void py_tns_write(const std::string & path, Tensor inds, Tensor vals,
                         std::vector<int64_t> dims) {
  if (vals.scalar_type() == torch::kFloat32) {
    auto tt = coo_from_torch<float>(inds, vals, dims);
    sp::tns_write(tt, path);
  } else {
    auto tt = coo_from_torch<double>(inds, vals, dims);
    sp::tns_write(tt, path);
  }
}

static void py_bin_write(const std::string & path, Tensor inds, Tensor vals,
                         std::vector<int64_t> dims, int idx_bytes, int val_bytes) {
  if (vals.scalar_type() == torch::kFloat32) {
    auto tt = coo_from_torch<float>(inds, vals, dims);
    sp::bin_write(tt, path, idx_bytes, val_bytes);
  } else {
    auto tt = coo_from_torch<double>(inds, vals, dims);
    sp::bin_write(tt, path, idx_bytes, val_bytes);
  }
}

// ------------------------------------------------------------- coo utils

template <typename V>
static py::tuple py_coo_fix_t(Tensor inds, Tensor vals, std::vector<int64_t> dims,
                              bool dedup, bool compress) {
  auto tt = coo_from_torch<V>(inds, vals, dims);
  int64_t ndups = 0, nempty = 0;
  if (dedup) {
    std::vector<int> perm(tt.nmodes);
    for (int m = 0; m < tt.nmodes; ++m) perm[m] = m;
    sp::coo_sort(tt, perm.data());
    ndups = (int64_t)sp::coo_remove_dups(tt);
  }
  py::list indmaps;
  if (compress) {
    nempty = (int64_t)sp::coo_remove_empty(tt);
    for (int m = 0; m < tt.nmodes; ++m) {
      if (tt.indmap[m].empty()) {
        indmaps.append(py::none());
      } else {
        Tensor t = torch::empty({(int64_t)tt.indmap[m].size()}, torch::kInt64);
        std::copy(tt.indmap[m].begin(), tt.indmap[m].end(), t.data_ptr<int64_t>());
        indmaps.append(t);
      }
    }
  }
  auto [i, v, d] = coo_to_torch(tt);
  return py::make_tuple(i, v, d, ndups, nempty, indmaps);
}

static py::tuple py_coo_fix(Tensor inds, Tensor vals, std::vector<int64_t> dims,
                            bool dedup, bool compress) {
  if (vals.scalar_type() == torch::kFloat32)
    return py_coo_fix_t<float>(inds, vals, dims, dedup, compress);
  return py_coo_fix_t<double>(inds, vals, dims, dedup, compress);
}

// ------------------------------------------------------------- csf build

template <typename V>
static py::object py_csf_build_t(Tensor inds, Tensor vals,
                                 std::vector<int64_t> dims,
                                 std::vector<int64_t> perm) {
  auto tt = coo_from_torch<V>(inds, vals, dims);
  int p[sp::MAX_NMODES];
  for (size_t i = 0; i < perm.size(); ++i) p[i] = (int)perm[i];
  auto c = sp::csf_build(tt, p);
  return csf_to_py(c);
}

static py::object py_csf_build(Tensor inds, Tensor vals,
                               std::vector<int64_t> dims,
                               std::vector<int64_t> perm) {
  if (vals.scalar_type() == torch::kFloat32)
    return py_csf_build_t<float>(inds, vals, dims, perm);
  return py_csf_build_t<double>(inds, vals, dims, perm);
}

// ---------------------------------------------------------------- mttkrp

template <typename V>
static Tensor py_mttkrp_stream_t(Tensor inds, Tensor vals,
                                 std::vector<int64_t> dims,
                                 std::vector<Tensor> mats, int mode) {
  auto tt = coo_from_torch<V>(inds, vals, dims);
  const int F = (int)mats[0].size(1);
  std::vector<const V*> mp;
  std::vector<Tensor> mc;
  for (auto & m : mats) { mc.push_back(m.contiguous()); mp.push_back(mc.back().data_ptr<V>()); }
  Tensor out = torch::empty({dims[mode], F}, mats[0].options());
  sp::mttkrp_stream(tt, mp.data(), out.data_ptr<V>(), mode, F);
  return out;
}

static Tensor py_mttkrp_stream(Tensor inds, Tensor vals, std::vector<int64_t> dims,
                               std::vector<Tensor> mats, int mode) {
  if (vals.scalar_type() == torch::kFloat32)
    return py_mttkrp_stream_t<float>(inds, vals, dims, mats, mode);
  return py_mttkrp_stream_t<double>(inds, vals, dims, mats, mode);
}

template <typename V>
static Tensor py_mttkrp_csf_t(py::dict csf, std::vector<Tensor> mats, int mode,
                              int nthreads) {
  auto c = csf_from_py<V>(csf);
  const int F = (int)mats[0].size(1);
  std::vector<const V*> mp;
  std::vector<Tensor> mc;
  for (auto & m : mats) { mc.push_back(m.contiguous()); mp.push_back(mc.back().data_ptr<V>()); }
  Tensor out = torch::empty({(int64_t)c.dims[mode], F}, mats[0].options());
  sp::mttkrp_csf_cpu(c, mp.data(), out.data_ptr<V>(), mode, F, nthreads);
  return out;
}

static Tensor py_mttkrp_csf(py::dict csf, std::vector<Tensor> mats, int mode,
                            int nthreads) {
  if (mats[0].scalar_type() == torch::kFloat32)
    return py_mttkrp_csf_t<float>(csf, mats, mode, nthreads);
  return py_mttkrp_csf_t<double>(csf, mats, mode, nthreads);
}

// ------------------------------------------------------------------- cpd

template <typename V>
static py::dict py_cpd_als_t(Tensor inds, Tensor vals, std::vector<int64_t> dims,
                             int rank, py::dict opts) {
  auto o = opts_from_dict(opts);
  auto tt = coo_from_torch<V>(inds, vals, dims);
  auto set = sp::csf_alloc(tt, o);
  auto k = sp::cpd_als(set, rank, o);
  py::dict r;
  py::list factors;
  const auto vopt = torch::TensorOptions().dtype(
      std::is_same<V, double>::value ? torch::kFloat64 : torch::kFloat32);
  for (int m = 0; m < k.nmodes; ++m) {
    Tensor t = torch::empty({(int64_t)k.dims[m], rank}, vopt);
    std::copy(k.factors[m].begin(), k.factors[m].end(), t.data_ptr<V>());
    factors.append(t);
  }
  Tensor lam = torch::empty({rank}, vopt);
  std::copy(k.lambda.begin(), k.lambda.end(), lam.data_ptr<V>());
  r["factors"] = factors;
  r["lambda"] = lam;
  r["fit"] = k.fit;
  r["niters"] = k.niters;
  return r;
}

static py::dict py_cpd_als(Tensor inds, Tensor vals, std::vector<int64_t> dims,
                           int rank, py::dict opts) {
  if (vals.scalar_type() == torch::kFloat32)
    return py_cpd_als_t<float>(inds, vals, dims, rank, opts);
  return py_cpd_als_t<double>(inds, vals, dims, rank, opts);
}

// --------------------------------------------------------------- dense ops

static Tensor py_seeded_init(int64_t nrows, int rank, int64_t row0,
                             uint64_t seed, int mode, const std::string & dtype) {
  if (dtype == "f32") {
    Tensor t = torch::empty({nrows, rank}, torch::kFloat32);
    sp::seeded_factor_init(t.data_ptr<float>(), (sp::idx_t)nrows, rank,
                           (sp::idx_t)row0, seed, mode);
    return t;
  }
  Tensor t = torch::empty({nrows, rank}, torch::kFloat64);
  sp::seeded_factor_init(t.data_ptr<double>(), (sp::idx_t)nrows, rank,
                         (sp::idx_t)row0, seed, mode);
  return t;
}

static std::vector<int64_t> py_order_modes(std::vector<int64_t> dims,
                                           const std::string & policy, int mode) {
  const int nm = (int)dims.size();
  sp::idx_t d[sp::MAX_NMODES];
  for (int m = 0; m < nm; ++m) d[m] = (sp::idx_t)dims[m];
  int p[sp::MAX_NMODES];
  if (policy == "smallfirst") sp::order_smallfirst(d, nm, p);
  else if (policy == "root") sp::order_root(d, nm, mode, p);
  else sp::order_leaf(d, nm, mode, p);
  return std::vector<int64_t>(p, p + nm);
}

// --------------------------------------------------- HIP launcher externs

extern "C" {
// defined in csrc/hip/mttkrp_kernels.hip (gfx950); stream is hipStream_t
void splatt_hip_mttkrp_root3_f64(
    const int64_t* fptr0, const int32_t* fids0, const int64_t* fptr1,
    const int32_t* fids1, const int32_t* fids2, const double* vals,
    int64_t nslices, int64_t nfibs, int64_t nnz,
    const double* A1, const double* A2, double* out, int rank, void* stream);
void splatt_hip_mttkrp_root3_f32(
    const int64_t* fptr0, const int32_t* fids0, const int64_t* fptr1,
    const int32_t* fids1, const int32_t* fids2, const float* vals,
    int64_t nslices, int64_t nfibs, int64_t nnz,
    const float* A1, const float* A2, float* out, int rank, void* stream);
void splatt_hip_mttkrp_intl3_f64(
    const int64_t* fptr0, const int32_t* fids0, const int64_t* fptr1,
    const int32_t* fids1, const int32_t* fids2, const double* vals,
    int64_t nslices, int64_t nfibs, int64_t nnz,
    const double* A0, const double* A2, double* out, int rank, void* stream);
void splatt_hip_mttkrp_intl3_f32(
    const int64_t* fptr0, const int32_t* fids0, const int64_t* fptr1,
    const int32_t* fids1, const int32_t* fids2, const float* vals,
    int64_t nslices, int64_t nfibs, int64_t nnz,
    const float* A0, const float* A2, float* out, int rank, void* stream);
void splatt_hip_mttkrp_leaf3_f64(
    const int64_t* fptr0, const int32_t* fids0, const int64_t* fptr1,
    const int32_t* fids1, const int32_t* fids2, const double* vals,
    int64_t nslices, int64_t nfibs, int64_t nnz,
    const double* A0, const double* A1, double* out, int rank, void* stream);
void splatt_hip_mttkrp_leaf3_f32(
    const int64_t* fptr0, const int32_t* fids0, const int64_t* fptr1,
    const int32_t* fids1, const int32_t* fids2, const float* vals,
    int64_t nslices, int64_t nfibs, int64_t nnz,
    const float* A0, const float* A1, float* out, int rank, void* stream);
int splatt_hip_kernels_arch(void);
// flat (expanded-CSF) kernels, csrc/hip/mttkrp_flat.hip
void splatt_hip_mttkrp_flat_f64(
    const int32_t*, const int32_t* const*, const double* const*,
    const double*, int64_t, double*, int, int, void*);
void splatt_hip_mttkrp_flat_f32(
    const int32_t*, const int32_t* const*, const float* const*,
    const float*, int64_t, float*, int, int, void*);
// deterministic flat kernel, csrc/hip/mttkrp_det.hip
int64_t splatt_hip_flat_det_ws(int64_t, int);
int splatt_hip_mttkrp_flat_det_f64(
    const int32_t*, const int32_t* const*, const double* const*,
    const double*, int64_t, double*, double*, int64_t, int, int, void*);
int splatt_hip_mttkrp_flat_det_f32(
    const int32_t*, const int32_t* const*, const float* const*,
    const float*, int64_t, float*, float*, int64_t, int, int, void*);
// dense kernels, csrc/hip/dense_kernels.hip
int splatt_hip_rowsolve_f64(const double*, const double*, double*, int64_t,
                            int, void*);
void splatt_hip_gram_det_f64(const double*, int64_t, int, double*, int64_t,
                             double*, void*);
void splatt_hip_gram_det_f32(const float*, int64_t, int, float*, int64_t,
                             float*, void*);
int splatt_hip_rowsolve_f32(const float*, const float*, float*, int64_t,
                            int, void*);
void splatt_hip_gram_f64(const double*, int64_t, int, double*, void*);
void splatt_hip_gram_f32(const float*, int64_t, int, float*, void*);
void splatt_hip_spd_inverse_f64(const double*, double*, int, void*);
void splatt_hip_spd_inverse_f32(const float*, float*, int, void*);
// LDS-staged flat kernel, csrc/hip/mttkrp_lds.hip
void splatt_hip_mttkrp_flat5_f64(
    const int32_t*, const int32_t*, const int32_t*, const int32_t*,
    const int32_t*, const double*, const double*, const double*,
    const double*, const double*, const int64_t*, const int64_t*,
    const int32_t*, int64_t, int32_t, int32_t, double*, int, int, void*);
void splatt_hip_mttkrp_flat5_f32(
    const int32_t*, const int32_t*, const int32_t*, const int32_t*,
    const int32_t*, const float*, const float*, const float*,
    const float*, const float*, const int64_t*, const int64_t*,
    const int32_t*, int64_t, int32_t, int32_t, float*, int, int, void*);
void splatt_hip_mttkrp_flat6_f64(
    const int32_t*, const double*, const double*, const double*,
    const double*, const int64_t*, const int64_t*, const int32_t*, int64_t,
    int32_t, int32_t, double*, int, int, void*);
void splatt_hip_mttkrp_flat6_f32(
    const int32_t*, const float*, const float*, const float*,
    const float*, const int64_t*, const int64_t*, const int32_t*, int64_t,
    int32_t, int32_t, float*, int, int, void*);
int splatt_hip_mttkrp_flat_f64f32(
    const int32_t*, const int32_t* const*, const float* const*,
    const double*, int64_t, double*, int, int, void*);
int splatt_hip_mttkrp_flat_f64bf16(
    const int32_t*, const int32_t* const*, const uint16_t* const*,
    const double*, int64_t, double*, int, int, void*);
void splatt_hip_mttkrp_flat6_f64f32(
    const int32_t*, const float*, const float*, const float*,
    const double*, const int64_t*, const int64_t*, const int32_t*, int64_t,
    int32_t, int32_t, double*, int, int, void*);
void splatt_hip_mttkrp_flat6_f64bf16(
    const int32_t*, const uint16_t*, const uint16_t*, const uint16_t*,
    const double*, const int64_t*, const int64_t*, const int32_t*, int64_t,
    int32_t, int32_t, double*, int, int, void*);
}

// LDS-staged variant: idx[0]/mats[0] is the bucketed level; block
// descriptors carry (nnz range, bucket row0)
static void py_gpu_mttkrp_flat5(Tensor key, std::vector<Tensor> idx,
                                std::vector<Tensor> mats, Tensor vals,
                                Tensor blk_start, Tensor blk_end,
                                Tensor blk_row0, int64_t chunk, int64_t dim0,
                                Tensor out, int64_t stream) {
  const int nother = (int)idx.size();
  TORCH_CHECK(nother >= 2 && nother <= 4);
  const int rank = (int)mats[0].size(1);
  const int32_t * ip[4] = {nullptr, nullptr, nullptr, nullptr};
  for (int t = 0; t < nother; ++t) ip[t] = idx[t].data_ptr<int32_t>();
  const int64_t nblocks = blk_start.numel();
  if (vals.scalar_type() == torch::kFloat64) {
    const double * mp[4] = {nullptr, nullptr, nullptr, nullptr};
    for (int t = 0; t < nother; ++t) mp[t] = mats[t].data_ptr<double>();
    splatt_hip_mttkrp_flat5_f64(key.data_ptr<int32_t>(), ip[0], ip[1], ip[2],
                                ip[3], mp[0], mp[1], mp[2], mp[3],
                                vals.data_ptr<double>(),
                                blk_start.data_ptr<int64_t>(),
                                blk_end.data_ptr<int64_t>(),
                                blk_row0.data_ptr<int32_t>(), nblocks,
                                (int32_t)chunk, (int32_t)dim0,
                                out.data_ptr<double>(), rank, nother,
                                (void*)stream);
  } else {
    const float * mp[4] = {nullptr, nullptr, nullptr, nullptr};
    for (int t = 0; t < nother; ++t) mp[t] = mats[t].data_ptr<float>();
    splatt_hip_mttkrp_flat5_f32(key.data_ptr<int32_t>(), ip[0], ip[1], ip[2],
                                ip[3], mp[0], mp[1], mp[2], mp[3],
                                vals.data_ptr<float>(),
                                blk_start.data_ptr<int64_t>(),
                                blk_end.data_ptr<int64_t>(),
                                blk_row0.data_ptr<int32_t>(), nblocks,
                                (int32_t)chunk, (int32_t)dim0,
                                out.data_ptr<float>(), rank, nother,
                                (void*)stream);
  }
}

extern "C" {
int splatt_hip_mttkrp_det6_f64(
    const int32_t*, const double*, const double*, const double*,
    const double*, const int64_t*, const int64_t*, const int32_t*,
    const int64_t*, int64_t,
    int32_t, int32_t, int64_t, int64_t, double*, double*, double*, int,
    int, void*);
int splatt_hip_mttkrp_det6_f32(
    const int32_t*, const float*, const float*, const float*,
    const float*, const int64_t*, const int64_t*, const int32_t*,
    const int64_t*, int64_t,
    int32_t, int32_t, int64_t, int64_t, float*, float*, float*, int, int,
    void*);
}

// LDS-staged bitwise-deterministic MTTKRP over the packed stream:
// per-bucket privatized outputs + ordered fixup + ascending-bucket fold
static void py_gpu_mttkrp_det6(Tensor pack, std::vector<Tensor> mats,
                               Tensor vals, Tensor blk_start,
                               Tensor blk_end, Tensor blk_row0,
                               Tensor blk_bucket_p0,
                               int64_t chunk, int64_t dim0,
                               int64_t nbuckets, Tensor outb, Tensor side,
                               Tensor out, int64_t stream) {
  const int nother = (int)mats.size();
  TORCH_CHECK(nother >= 2 && nother <= 3);
  TORCH_CHECK(pack.is_contiguous() && pack.size(1) == 4
              && pack.scalar_type() == torch::kInt32);
  const int rank = (int)mats[0].size(1);
  const int64_t nblocks = blk_start.numel();
  const int64_t nrows_out = out.size(0);
  int rc;
  if (vals.scalar_type() == torch::kFloat64) {
    const double * mp[3] = {nullptr, nullptr, nullptr};
    for (int t = 0; t < nother; ++t) mp[t] = mats[t].data_ptr<double>();
    rc = splatt_hip_mttkrp_det6_f64(
        pack.data_ptr<int32_t>(), mp[0], mp[1], mp[2],
        vals.data_ptr<double>(), blk_start.data_ptr<int64_t>(),
        blk_end.data_ptr<int64_t>(), blk_row0.data_ptr<int32_t>(),
        blk_bucket_p0.data_ptr<int64_t>(), nblocks,
        (int32_t)chunk, (int32_t)dim0, nrows_out, nbuckets,
        outb.data_ptr<double>(), side.data_ptr<double>(),
        out.data_ptr<double>(), rank, nother, (void*)stream);
  } else {
    const float * mp[3] = {nullptr, nullptr, nullptr};
    for (int t = 0; t < nother; ++t) mp[t] = mats[t].data_ptr<float>();
    rc = splatt_hip_mttkrp_det6_f32(
        pack.data_ptr<int32_t>(), mp[0], mp[1], mp[2],
        vals.data_ptr<float>(), blk_start.data_ptr<int64_t>(),
        blk_end.data_ptr<int64_t>(), blk_row0.data_ptr<int32_t>(),
        blk_bucket_p0.data_ptr<int64_t>(), nblocks,
        (int32_t)chunk, (int32_t)dim0, nrows_out, nbuckets,
        outb.data_ptr<float>(), side.data_ptr<float>(),
        out.data_ptr<float>(), rank, nother, (void*)stream);
  }
  TORCH_CHECK(rc == 0, "det6 supports spec ranks and 3-4 modes, rc=", rc);
}

// packed-stream LDS variant: one int4 word per nonzero
// (x = output key, y = staged level, z/w = remaining levels)
static void py_gpu_mttkrp_flat6(Tensor pack, std::vector<Tensor> mats,
                                Tensor vals, Tensor blk_start,
                                Tensor blk_end, Tensor blk_row0,
                                int64_t chunk, int64_t dim0, Tensor out,
                                int64_t stream) {
  const int nother = (int)mats.size();
  TORCH_CHECK(nother >= 2 && nother <= 3);
  TORCH_CHECK(pack.is_contiguous() && pack.size(1) == 4
              && pack.scalar_type() == torch::kInt32,
              "pack must be contiguous [nnz,4] int32");
  const int rank = (int)mats[0].size(1);
  const int64_t nblocks = blk_start.numel();
  if (vals.scalar_type() == torch::kFloat64) {
    // reduced-precision factor STORAGE with f64 accumulation
    if (mats[0].scalar_type() == torch::kFloat32) {
      const float * mp[3] = {nullptr, nullptr, nullptr};
      for (int t = 0; t < nother; ++t) mp[t] = mats[t].data_ptr<float>();
      splatt_hip_mttkrp_flat6_f64f32(pack.data_ptr<int32_t>(), mp[0], mp[1],
                                     mp[2], vals.data_ptr<double>(),
                                     blk_start.data_ptr<int64_t>(),
                                     blk_end.data_ptr<int64_t>(),
                                     blk_row0.data_ptr<int32_t>(), nblocks,
                                     (int32_t)chunk, (int32_t)dim0,
                                     out.data_ptr<double>(), rank, nother,
                                     (void*)stream);
      return;
    }
    if (mats[0].scalar_type() == torch::kBFloat16) {
      const uint16_t * mp[3] = {nullptr, nullptr, nullptr};
      for (int t = 0; t < nother; ++t)
        mp[t] = reinterpret_cast<const uint16_t*>(
            mats[t].data_ptr<at::BFloat16>());
      splatt_hip_mttkrp_flat6_f64bf16(pack.data_ptr<int32_t>(), mp[0],
                                      mp[1], mp[2], vals.data_ptr<double>(),
                                      blk_start.data_ptr<int64_t>(),
                                      blk_end.data_ptr<int64_t>(),
                                      blk_row0.data_ptr<int32_t>(), nblocks,
                                      (int32_t)chunk, (int32_t)dim0,
                                      out.data_ptr<double>(), rank, nother,
                                      (void*)stream);
      return;
    }
    const double * mp[3] = {nullptr, nullptr, nullptr};
    for (int t = 0; t < nother; ++t) mp[t] = mats[t].data_ptr<double>();
    splatt_hip_mttkrp_flat6_f64(pack.data_ptr<int32_t>(), mp[0], mp[1],
                                mp[2], vals.data_ptr<double>(),
                                blk_start.data_ptr<int64_t>(),
                                blk_end.data_ptr<int64_t>(),
                                blk_row0.data_ptr<int32_t>(), nblocks,
                                (int32_t)chunk, (int32_t)dim0,
                                out.data_ptr<double>(), rank, nother,
                                (void*)stream);
  } else {
    const float * mp[3] = {nullptr, nullptr, nullptr};
    for (int t = 0; t < nother; ++t) mp[t] = mats[t].data_ptr<float>();
    splatt_hip_mttkrp_flat6_f32(pack.data_ptr<int32_t>(), mp[0], mp[1],
                                mp[2], vals.data_ptr<float>(),
                                blk_start.data_ptr<int64_t>(),
                                blk_end.data_ptr<int64_t>(),
                                blk_row0.data_ptr<int32_t>(), nblocks,
                                (int32_t)chunk, (int32_t)dim0,
                                out.data_ptr<float>(), rank, nother,
                                (void*)stream);
  }
}

// G (FxF, pre-zeroed) += A^T A for tall-skinny device A
static void py_gpu_gram(Tensor A, Tensor G, int64_t stream) {
  const int64_t n = A.size(0);
  const int F = (int)A.size(1);
  if (A.scalar_type() == torch::kFloat64)
    splatt_hip_gram_f64(A.data_ptr<double>(), n, F, G.data_ptr<double>(),
                        (void*)stream);
  else
    splatt_hip_gram_f32(A.data_ptr<float>(), n, F, G.data_ptr<float>(),
                        (void*)stream);
}

// key + per-other-level (idx, mat) pairs; nnz products folded by key runs
static void py_gpu_mttkrp_flat(Tensor key, std::vector<Tensor> idx,
                               std::vector<Tensor> mats, Tensor vals,
                               Tensor out, int64_t stream) {
  const int nother = (int)idx.size();
  TORCH_CHECK(nother >= 2 && nother <= 7, "flat kernel supports 3..8 modes");
  TORCH_CHECK((int)mats.size() == nother);
  const int rank = (int)mats[0].size(1);
  const int64_t nnz = vals.numel();
  const int32_t * ip[8] = {};
  for (int t = 0; t < nother; ++t) ip[t] = idx[t].data_ptr<int32_t>();
  if (vals.scalar_type() == torch::kFloat64) {
    // reduced-precision factor STORAGE with f64 accumulation (v2 path)
    if (mats[0].scalar_type() == torch::kFloat32) {
      const float * mp[8] = {};
      for (int t = 0; t < nother; ++t) mp[t] = mats[t].data_ptr<float>();
      TORCH_CHECK(splatt_hip_mttkrp_flat_f64f32(
                      key.data_ptr<int32_t>(), ip, mp,
                      vals.data_ptr<double>(), nnz, out.data_ptr<double>(),
                      rank, nother, (void*)stream) == 0,
                  "f32 factor store needs rank in {4,8,16,32,64}");
      return;
    }
    if (mats[0].scalar_type() == torch::kBFloat16) {
      const uint16_t * mp[8] = {};
      for (int t = 0; t < nother; ++t)
        mp[t] = reinterpret_cast<const uint16_t*>(
            mats[t].data_ptr<at::BFloat16>());
      TORCH_CHECK(splatt_hip_mttkrp_flat_f64bf16(
                      key.data_ptr<int32_t>(), ip, mp,
                      vals.data_ptr<double>(), nnz, out.data_ptr<double>(),
                      rank, nother, (void*)stream) == 0,
                  "bf16 factor store needs rank in {4,8,16,32,64}");
      return;
    }
    const double * mp[8] = {};
    for (int t = 0; t < nother; ++t) mp[t] = mats[t].data_ptr<double>();
    splatt_hip_mttkrp_flat_f64(key.data_ptr<int32_t>(), ip, mp,
                               vals.data_ptr<double>(), nnz,
                               out.data_ptr<double>(), rank, nother,
                               (void*)stream);
  } else {
    const float * mp[8] = {};
    for (int t = 0; t < nother; ++t) mp[t] = mats[t].data_ptr<float>();
    splatt_hip_mttkrp_flat_f32(key.data_ptr<int32_t>(), ip, mp,
                               vals.data_ptr<float>(), nnz,
                               out.data_ptr<float>(), rank, nother,
                               (void*)stream);
  }
}

// deterministic flat MTTKRP (csrc/hip/mttkrp_det.hip): plain stores for
// walker-interior key runs + ordered fixup of boundary partials in `side`
static void py_gpu_mttkrp_flat_det(Tensor key, std::vector<Tensor> idx,
                                   std::vector<Tensor> mats, Tensor vals,
                                   Tensor out, Tensor side, int64_t stream) {
  const int nother = (int)idx.size();
  TORCH_CHECK(nother >= 2 && nother <= 4,
              "deterministic kernel supports 3..5 modes");
  TORCH_CHECK((int)mats.size() == nother);
  const int rank = (int)mats[0].size(1);
  const int64_t nnz = vals.numel();
  const int32_t * ip[8] = {};
  for (int t = 0; t < nother; ++t) ip[t] = idx[t].data_ptr<int32_t>();
  int rc;
  if (vals.scalar_type() == torch::kFloat64) {
    const double * mp[8] = {};
    for (int t = 0; t < nother; ++t) mp[t] = mats[t].data_ptr<double>();
    rc = splatt_hip_mttkrp_flat_det_f64(key.data_ptr<int32_t>(), ip, mp,
                                        vals.data_ptr<double>(), nnz,
                                        out.data_ptr<double>(),
                                        side.data_ptr<double>(),
                                        side.numel(), rank, nother,
                                        (void*)stream);
  } else {
    const float * mp[8] = {};
    for (int t = 0; t < nother; ++t) mp[t] = mats[t].data_ptr<float>();
    rc = splatt_hip_mttkrp_flat_det_f32(key.data_ptr<int32_t>(), ip, mp,
                                        vals.data_ptr<float>(), nnz,
                                        out.data_ptr<float>(),
                                        side.data_ptr<float>(),
                                        side.numel(), rank, nother,
                                        (void*)stream);
  }
  TORCH_CHECK(rc != -1, "deterministic kernel requires rank in "
                        "{4,8,16,32,64} and <= 5 modes");
  TORCH_CHECK(rc == 0, "deterministic workspace too small (", side.numel(),
              " elems < ", splatt_hip_flat_det_ws(nnz, rank), ")");
}

// thin wrappers: Python passes contiguous CUDA tensors + stream handle
template <typename V>
static void gpu_mttkrp3_t(int which, Tensor fptr0, py::object fids0, Tensor fptr1,
                          Tensor fids1, Tensor fids2, Tensor vals,
                          Tensor Ma, Tensor Mb, Tensor out, int64_t stream) {
  const int64_t nslices = fptr0.numel() - 1;
  const int64_t nfibs = fptr1.numel() - 1;
  const int64_t nnz = vals.numel();
  const int rank = (int)Ma.size(1);
  const int32_t * f0 = fids0.is_none() ? nullptr
                       : fids0.cast<Tensor>().data_ptr<int32_t>();
  const bool f64 = std::is_same<V, double>::value;
  auto fp0 = fptr0.data_ptr<int64_t>();
  auto fp1 = fptr1.data_ptr<int64_t>();
  auto fi1 = fids1.data_ptr<int32_t>();
  auto fi2 = fids2.data_ptr<int32_t>();
  auto vp = vals.data_ptr<V>();
  auto a = Ma.data_ptr<V>();
  auto b = Mb.data_ptr<V>();
  auto o = out.data_ptr<V>();
  void * s = (void*)stream;
  if (f64) {
    auto vd = (const double*)vp; auto ad = (const double*)a;
    auto bd = (const double*)b; auto od = (double*)o;
    if (which == 0) splatt_hip_mttkrp_root3_f64(fp0, f0, fp1, fi1, fi2, vd, nslices, nfibs, nnz, ad, bd, od, rank, s);
    else if (which == 1) splatt_hip_mttkrp_intl3_f64(fp0, f0, fp1, fi1, fi2, vd, nslices, nfibs, nnz, ad, bd, od, rank, s);
    else splatt_hip_mttkrp_leaf3_f64(fp0, f0, fp1, fi1, fi2, vd, nslices, nfibs, nnz, ad, bd, od, rank, s);
  } else {
    auto vf = (const float*)vp; auto af = (const float*)a;
    auto bf = (const float*)b; auto of = (float*)o;
    if (which == 0) splatt_hip_mttkrp_root3_f32(fp0, f0, fp1, fi1, fi2, vf, nslices, nfibs, nnz, af, bf, of, rank, s);
    else if (which == 1) splatt_hip_mttkrp_intl3_f32(fp0, f0, fp1, fi1, fi2, vf, nslices, nfibs, nnz, af, bf, of, rank, s);
    else splatt_hip_mttkrp_leaf3_f32(fp0, f0, fp1, fi1, fi2, vf, nslices, nfibs, nnz, af, bf, of, rank, s);
  }
}

static void py_gpu_mttkrp3(int which, Tensor fptr0, py::object fids0, Tensor fptr1,
                           Tensor fids1, Tensor fids2, Tensor vals,
                           Tensor Ma, Tensor Mb, Tensor out, int64_t stream) {
  if (vals.scalar_type() == torch::kFloat32)
    gpu_mttkrp3_t<float>(which, fptr0, fids0, fptr1, fids1, fids2, vals, Ma, Mb, out, stream);
  else
    gpu_mttkrp3_t<double>(which, fptr0, fids0, fptr1, fids1, fids2, vals, Ma, Mb, out, stream);
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("tensor_load", &py_tensor_load, "load .tns/.bin tensor");
  m.def("tns_write", &py_tns_write);
  m.def("bin_write", &py_bin_write);
  m.def("coo_fix", &py_coo_fix, "sort+dedup and/or remove empty slices");
  m.def("csf_build", &py_csf_build, "CPU CSF build for a level permutation");
  m.def("mttkrp_stream", &py_mttkrp_stream, "COO gold-oracle MTTKRP (CPU)");
  m.def("mttkrp_csf_cpu", &py_mttkrp_csf, "CSF MTTKRP (CPU)");
  m.def("cpd_als_cpu", &py_cpd_als, "CPD-ALS on the CPU reference path");
  m.def("seeded_init", &py_seeded_init, "partition-invariant seeded factor init");
  m.def("order_modes", &py_order_modes, "CSF mode-order policies");
  m.def("gpu_mttkrp3", &py_gpu_mttkrp3, "3-mode CSF MTTKRP HIP kernels");
  m.def("gpu_mttkrp_flat", &py_gpu_mttkrp_flat,
        "flat expanded-CSF MTTKRP HIP kernel (3..5 modes)");
  m.def("gpu_mttkrp_flat_det", &py_gpu_mttkrp_flat_det,
        "bitwise-deterministic flat MTTKRP (depth-0 streams, spec ranks)");
  m.def("flat_det_ws_elems", &splatt_hip_flat_det_ws,
        "workspace elements required by gpu_mttkrp_flat_det");
  m.def("gpu_gram", &py_gpu_gram, "G += A^T A (tall-skinny, F<=64)");
  m.def("gpu_gram_det", [](Tensor A, Tensor Gpart, Tensor G,
                           int64_t stream) {
    const int64_t n = A.size(0);
    const int F = (int)A.size(1);
    const int64_t nparts = Gpart.numel() / (F * F);
    TORCH_CHECK(nparts >= 1, "gram_det workspace too small");
    if (A.scalar_type() == torch::kFloat64)
      splatt_hip_gram_det_f64(A.data_ptr<double>(), n, F,
                              Gpart.data_ptr<double>(), nparts,
                              G.data_ptr<double>(), (void*)stream);
    else
      splatt_hip_gram_det_f32(A.data_ptr<float>(), n, F,
                              Gpart.data_ptr<float>(), nparts,
                              G.data_ptr<float>(), (void*)stream);
  }, "G = A^T A, per-block partials folded in fixed order "
     "(bitwise-deterministic)");
  m.def("gpu_rowsolve", [](Tensor A, Tensor B, Tensor C, int64_t stream) {
    const int64_t n = A.size(0);
    const int F = (int)A.size(1);
    int rc;
    if (A.scalar_type() == torch::kFloat64)
      rc = splatt_hip_rowsolve_f64(A.data_ptr<double>(), B.data_ptr<double>(),
                                   C.data_ptr<double>(), n, F, (void*)stream);
    else
      rc = splatt_hip_rowsolve_f32(A.data_ptr<float>(), B.data_ptr<float>(),
                                   C.data_ptr<float>(), n, F, (void*)stream);
    TORCH_CHECK(rc == 0, "rowsolve supports F in {4,8,16,32,64}, got ", F);
  }, "C = A @ B (B is FxF, staged in LDS; bitwise-deterministic)");
  m.def("gpu_spd_inverse", [](Tensor G, Tensor Ginv, int64_t stream) {
    const int F = (int)G.size(0);
    if (G.scalar_type() == torch::kFloat64)
      splatt_hip_spd_inverse_f64(G.data_ptr<double>(), Ginv.data_ptr<double>(),
                                 F, (void*)stream);
    else
      splatt_hip_spd_inverse_f32(G.data_ptr<float>(), Ginv.data_ptr<float>(),
                                 F, (void*)stream);
  }, "Ginv = G^-1 for SPD FxF (F<=64), one-workgroup Cholesky");
  m.def("gpu_mttkrp_flat5", &py_gpu_mttkrp_flat5,
        "LDS-staged flat MTTKRP (bucketed builds, root output)");
  m.def("gpu_mttkrp_flat6", &py_gpu_mttkrp_flat6,
        "LDS-staged flat MTTKRP over the packed int4 stream");
  m.def("gpu_mttkrp_det6", &py_gpu_mttkrp_det6,
        "LDS-staged deterministic MTTKRP (bucket-privatized outputs)");
  m.def("partition_weighted", [](std::vector<int64_t> w, int nparts) {
    int64_t bn = 0;
    auto parts = sp::partition_weighted(w.data(), (int64_t)w.size(), nparts, &bn);
    return py::make_tuple(parts, bn);
  }, "optimal chains-on-chains partition (boundaries, bottleneck)");
  m.def("hip_arch", []() { return splatt_hip_kernels_arch(); });
}
