// Core types for the MI355X-native sparse tensor factorization engine.
//
// Capability parity target: SPLATT's type/config layer
// (reference: include/splatt/types_config.h:36-215, constants.h:14-19).
// Design departure: the core is templated C++17 (value type float/double),
// indices are fixed 64-bit at the API with 32-bit node ids inside CSF levels
// (device bandwidth), instead of compile-time-selected widths.
#pragma once

#include <cstdint>
#include <cstddef>
#include <vector>
#include <array>
#include <string>
#include <stdexcept>

namespace splatt {

using idx_t = uint64_t;   // external index type (matches splatt_idx_t u64 default)
using fid_t = uint32_t;   // per-level node id inside a CSF shard (dims < 2^32)

constexpr int MAX_NMODES = 8;

enum class CsfAlloc { ONEMODE, TWOMODE, ALLMODE };
enum class TileMode { NOTILE, DENSETILE };

enum ErrorCode {
  SPLATT_OK = 0,
  SPLATT_ERR_BADINPUT = -1,
  SPLATT_ERR_NOMEMORY = -2,
};

struct Options {
  double tolerance   = 1e-5;
  idx_t  max_iters   = 50;
  CsfAlloc csf_alloc = CsfAlloc::TWOMODE;
  TileMode tile      = TileMode::NOTILE;
  idx_t  tile_depth  = 1;       // number of leaf-side levels tiled
  double privatize_threshold = 0.02;
  double regularize  = 0.0;     // ridge term added to the Gram diagonal
  uint64_t seed      = 0;       // 0 -> nondeterministic
  int    nthreads    = 0;       // 0 -> omp default
  int    verbosity   = 1;
};

// 64-byte aligned allocation helper (mirrors splatt_malloc semantics,
// reference src/base.c:42).
void * aligned_alloc64(size_t bytes);
void aligned_free64(void * ptr);

template <typename T>
struct AlignedDeleter { void operator()(T * p) const { aligned_free64((void*)p); } };

}  // namespace splatt
