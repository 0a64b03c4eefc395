// `splatt` CLI binary (native, torch-free).
// Parity: reference cmds/splatt_bin.c:76-123 dispatcher with sub-commands
// cpd / check / convert / stats / bench / reorder (cmds/cmd_*.c). The
// Python CLI (`python -m splatt_amd`) is the full-featured front end
// (GPU paths, graph exports, graph/hgraph-driven reorders); this binary
// covers the host library surface.
#include <cstdio>
#include <cstring>
#include <string>
#include <array>
#include <vector>
#include <chrono>

#include "splatt.h"
#include "../core/types.hpp"
#include "../core/sptensor.hpp"
#include "../core/csf.hpp"
#include "../core/mttkrp_cpu.hpp"
#include "../core/cpd.hpp"
#include "../core/io.hpp"

using namespace splatt;

static void usage() {
  std::printf(
      "splatt — MI355X-native sparse tensor factorization (host CLI)\n\n"
      "  splatt cpd TENSOR [-r RANK] [-i ITERS] [-t TOL] [--seed S]"
      " [--reg R] [--nowrite]\n"
      "  splatt check TENSOR [--fix OUT]\n"
      "  splatt convert TENSOR OUT          (.bin <-> .tns by extension)\n"
      "  splatt stats TENSOR\n"
      "  splatt bench TENSOR [-r RANK] [-N ITERS]\n"
      "  splatt reorder TENSOR OUT [--type rand|perm] [--seed S]"
      " [--permfile PREFIX]\n");
}

static double argf(int argc, char ** argv, const char * flag, double dflt) {
  for (int i = 0; i < argc - 1; ++i)
    if (!std::strcmp(argv[i], flag)) return std::atof(argv[i + 1]);
  return dflt;
}
static bool has_flag(int argc, char ** argv, const char * flag) {
  for (int i = 0; i < argc; ++i)
    if (!std::strcmp(argv[i], flag)) return true;
  return false;
}
static const char * args(int argc, char ** argv, const char * flag) {
  for (int i = 0; i < argc - 1; ++i)
    if (!std::strcmp(argv[i], flag)) return argv[i + 1];
  return nullptr;
}

static void print_stats(const SpTensor<double> & tt, const char * fname) {
  std::printf("Tensor information ---------------------------------\n");
  std::printf("FILE=%s\n", fname);
  std::printf("DIMS=");
  for (int m = 0; m < tt.nmodes; ++m)
    std::printf("%llu%s", (unsigned long long)tt.dims[m],
                m + 1 == tt.nmodes ? "" : "x");
  std::printf(" NNZ=%llu DENSITY=%.4e\n\n",
              (unsigned long long)tt.nnz, tt.density());
}

static int cmd_cpd(int argc, char ** argv) {
  const char * fname = argv[0];
  double * o = splatt_default_opts();
  o[SPLATT_OPTION_NITER] = argf(argc, argv, "-i", 50);
  o[SPLATT_OPTION_TOLERANCE] = argf(argc, argv, "-t", 1e-5);
  o[SPLATT_OPTION_RANDSEED] = argf(argc, argv, "--seed", (double)0x5EED5EEDull);
  o[SPLATT_OPTION_REGULARIZE] = argf(argc, argv, "--reg", 0.0);
  // GPU engine: root-output (ALLMODE) streams are 3-5x faster kernels;
  // keep the reference's TWOMODE default on the CPU path
  if (splatt_gpu_available())
    o[SPLATT_OPTION_CSF_ALLOC] = SPLATT_CSF_ALLMODE;
  const int rank = (int)argf(argc, argv, "-r", 10);

  splatt_idx_t nmodes = 0;
  splatt_csf * csf = nullptr;
  if (splatt_csf_load(fname, &nmodes, &csf, o) != SPLATT_SUCCESS) {
    std::fprintf(stderr, "error loading %s\n", fname);
    return 1;
  }
  auto t0 = std::chrono::steady_clock::now();
  splatt_kruskal k;
  if (splatt_cpd_als(csf, rank, o, &k) != SPLATT_SUCCESS) return 1;
  const double secs = std::chrono::duration<double>(
      std::chrono::steady_clock::now() - t0).count();
  std::printf("Final fit: %.5f   CPD time: %.3fs\n", k.fit, secs);
  if (!has_flag(argc, argv, "--nowrite")) {
    for (splatt_idx_t m = 0; m < k.nmodes; ++m) {
      char name[64];
      std::snprintf(name, sizeof name, "mode%llu.mat",
                    (unsigned long long)m + 1);
      mat_write(k.factors[m], k.dims[m], rank, name);
    }
    vec_write(k.lambda, (idx_t)rank, "lambda.mat");
  }
  splatt_free_kruskal(&k);
  splatt_free_csf(csf, o);
  splatt_free_opts(o);
  return 0;
}

static int cmd_check(int argc, char ** argv) {
  auto tt = tensor_load<double>(argv[0]);
  std::vector<int> perm(tt.nmodes);
  for (int m = 0; m < tt.nmodes; ++m) perm[m] = m;
  coo_sort(tt, perm.data());
  const idx_t dups = coo_remove_dups(tt);
  const idx_t empty = coo_remove_empty(tt);
  std::printf("duplicates merged: %llu; empty slices removed: %llu\n",
              (unsigned long long)dups, (unsigned long long)empty);
  if (const char * out = args(argc, argv, "--fix")) {
    tns_write(tt, out);
    std::printf("wrote %s\n", out);
  }
  return 0;
}

static int cmd_convert(int argc, char ** argv) {
  if (argc < 2) { usage(); return 1; }
  auto tt = tensor_load<double>(argv[0]);
  const std::string out = argv[1];
  if (out.size() > 4 && out.substr(out.size() - 4) == ".bin")
    bin_write(tt, out);
  else
    tns_write(tt, out);
  std::printf("wrote %s\n", out.c_str());
  return 0;
}

static int cmd_stats(int argc, char ** argv) {
  auto tt = tensor_load<double>(argv[0]);
  print_stats(tt, argv[0]);
  return 0;
}

static int cmd_reorder(int argc, char ** argv) {
  // splatt reorder TENSOR OUT [--type rand|perm] [--seed S]
  //                [--permfile PREFIX]
  // rand: fresh random permutation per mode; perm: read PREFIX.modeM.perm
  // (the Python CLI's file format; it also offers graph/hgraph-driven
  // orders). --permfile with rand writes the generated permutation.
  if (argc < 2) { usage(); return 1; }
  auto tt = tensor_load<double>(argv[0]);
  const char * type = args(argc, argv, "--type");
  const char * pfx = args(argc, argv, "--permfile");
  const uint64_t seed = (uint64_t)argf(argc, argv, "--seed", 42);
  const bool apply_file = type && !std::strcmp(type, "perm");
  if (apply_file && !pfx) {
    std::fprintf(stderr, "reorder --type perm needs --permfile\n");
    return 1;
  }
  std::array<std::vector<idx_t>, MAX_NMODES> perm;
  for (int m = 0; m < tt.nmodes; ++m) {
    perm[m].resize(tt.dims[m]);
    if (apply_file) {
      char name[512];
      std::snprintf(name, sizeof name, "%s.mode%d.perm", pfx, m);
      FILE * f = std::fopen(name, "r");
      if (!f) throw std::runtime_error(std::string("cannot open ") + name);
      unsigned long long v;
      for (idx_t i = 0; i < tt.dims[m]; ++i) {
        if (std::fscanf(f, "%llu", &v) != 1 || v >= tt.dims[m]) {
          std::fclose(f);
          throw std::runtime_error(std::string("bad perm file ") + name);
        }
        perm[m][i] = (idx_t)v;
      }
      std::fclose(f);
    } else {
      for (idx_t i = 0; i < tt.dims[m]; ++i) perm[m][i] = i;
      // Fisher-Yates with splitmix64-style mixing per mode
      uint64_t s = seed * 0x9E3779B97F4A7C15ull + (uint64_t)m + 1;
      for (idx_t i = tt.dims[m] - 1; i > 0; --i) {
        s += 0x9E3779B97F4A7C15ull;
        uint64_t z = s;
        z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
        z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
        z ^= z >> 31;
        std::swap(perm[m][i], perm[m][z % (i + 1)]);
      }
    }
  }
  for (int m = 0; m < tt.nmodes; ++m)
    for (idx_t i = 0; i < tt.nnz; ++i)
      tt.ind[m][i] = perm[m][tt.ind[m][i]];
  tns_write(tt, argv[1]);
  std::printf("wrote %s\n", argv[1]);
  if (pfx && !apply_file) {
    for (int m = 0; m < tt.nmodes; ++m) {
      char name[512];
      std::snprintf(name, sizeof name, "%s.mode%d.perm", pfx, m);
      FILE * f = std::fopen(name, "w");
      if (!f) throw std::runtime_error(std::string("cannot write ") + name);
      for (idx_t i = 0; i < tt.dims[m]; ++i)
        std::fprintf(f, "%llu\n", (unsigned long long)perm[m][i]);
      std::fclose(f);
    }
    std::printf("wrote %s.mode*.perm\n", pfx);
  }
  return 0;
}

static int cmd_bench(int argc, char ** argv) {
  auto tt = tensor_load<double>(argv[0]);
  print_stats(tt, argv[0]);
  const int rank = (int)argf(argc, argv, "-r", 16);
  const int iters = (int)argf(argc, argv, "-N", 3);
  Options opt;
  auto set = csf_alloc(tt, opt);
  std::array<std::vector<double>, MAX_NMODES> mats;
  std::array<const double*, MAX_NMODES> mp{};
  idx_t maxdim = 0;
  for (int m = 0; m < tt.nmodes; ++m) {
    mats[m].resize(tt.dims[m] * (idx_t)rank);
    seeded_factor_init(mats[m].data(), tt.dims[m], rank, 0, 123, m);
    mp[m] = mats[m].data();
    maxdim = std::max(maxdim, tt.dims[m]);
  }
  std::vector<double> out(maxdim * (idx_t)rank);
  for (int m = 0; m < tt.nmodes; ++m) {
    auto t0 = std::chrono::steady_clock::now();
    for (int i = 0; i < iters; ++i) {
      const auto & c = set.csfs[set.mode_csf[m]];
      mttkrp_csf_cpu(c, mp.data(), out.data(), m, rank);
    }
    const double secs = std::chrono::duration<double>(
        std::chrono::steady_clock::now() - t0).count() / iters;
    std::printf("  mode %d: %9.3f ms  (%.2f GFLOP/s)\n", m, secs * 1e3,
                3.0 * tt.nnz * rank / secs / 1e9);
  }
  return 0;
}

int main(int argc, char ** argv) {
  if (argc < 3) { usage(); return argc < 2 ? 0 : 1; }
  const std::string cmd = argv[1];
  int sub_argc = argc - 2;
  char ** sub_argv = argv + 2;
  try {
    if (cmd == "cpd") return cmd_cpd(sub_argc, sub_argv);
    if (cmd == "check") return cmd_check(sub_argc, sub_argv);
    if (cmd == "convert") return cmd_convert(sub_argc, sub_argv);
    if (cmd == "stats") return cmd_stats(sub_argc, sub_argv);
    if (cmd == "bench") return cmd_bench(sub_argc, sub_argv);
    if (cmd == "reorder") return cmd_reorder(sub_argc, sub_argv);
  } catch (const std::exception & e) {
    std::fprintf(stderr, "splatt: %s\n", e.what());
    return 1;
  }
  usage();
  return 1;
}
