// Tensor / matrix I/O. Capability parity: reference src/io.c
// (text .tns reader with 0/1-index autodetect io.c:62-348; binary .bin
// format with magic + index/value widths io.c:118-195/388-474; matrix and
// vector writers io.c:656-845). Fresh implementation: buffered C++ streams,
// same observable formats (.tns interchange-compatible with the reference).
#pragma once

#include "sptensor.hpp"
#include <string>

namespace splatt {

template <typename V>
SpTensor<V> tns_read(const std::string & path);

template <typename V>
void tns_write(const SpTensor<V> & tt, const std::string & path);

// Binary tensor format: little-endian header
//   magic "SPLATTB1" | u32 idx_bytes | u32 val_bytes | u64 nmodes | u64 nnz
//   | u64 dims[nmodes] | idx ind[m][nnz] per mode | val vals[nnz]
// Readers down/up-convert widths on load (reference io.c:477-555 semantics).
template <typename V>
SpTensor<V> bin_read(const std::string & path);

template <typename V>
void bin_write(const SpTensor<V> & tt, const std::string & path,
               int idx_bytes = 8, int val_bytes = sizeof(V));

// dispatch on extension (.tns/.coo text, .bin binary)
template <typename V>
SpTensor<V> tensor_load(const std::string & path);

template <typename V>
void mat_write(const V * A, idx_t nrows, int ncols, const std::string & path);

template <typename V>
void vec_write(const V * v, idx_t n, const std::string & path);

}  // namespace splatt
