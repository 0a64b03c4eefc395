#include "io.hpp"
#include <cstdio>
#include <memory>
#include <cstring>
#include <cctype>
#include <stdexcept>

namespace splatt {

namespace {

struct FileCloser { void operator()(FILE * f) const { if (f) fclose(f); } };
using FilePtr = std::unique_ptr<FILE, FileCloser>;

FilePtr xopen(const std::string & path, const char * mode) {
  FILE * f = fopen(path.c_str(), mode);
  if (!f) throw std::runtime_error("cannot open " + path);
  return FilePtr(f);
}

// parse one line of whitespace-separated numbers; returns token count.
// `clean` (optional) reports whether the whole line was consumed — false
// means a non-numeric token or more tokens than maxtok (malformed input,
// which a silent parser would truncate into wrong coordinates).
inline int parse_line(const char * s, double * out, int maxtok,
                      bool * clean = nullptr) {
  int n = 0;
  bool ok = true;
  while (*s) {
    while (*s && std::isspace((unsigned char)*s)) ++s;
    if (!*s || *s == '#' || *s == '%') break;
    if (n >= maxtok) { ok = false; break; }
    char * end = nullptr;
    out[n] = strtod(s, &end);
    if (end == s) { ok = false; break; }
    s = end;
    ++n;
  }
  if (clean) *clean = ok;
  return n;
}

}  // namespace

template <typename V>
SpTensor<V> tns_read(const std::string & path) {
  FilePtr f = xopen(path, "r");
  char * line = nullptr;
  size_t cap = 0;
  double tok[MAX_NMODES + 2];

  // pass 1: nmodes, nnz, dims, min index (0/1 autodetect)
  int nmodes = 0;
  idx_t nnz = 0;
  idx_t dims[MAX_NMODES] = {0};
  idx_t minidx = ~(idx_t)0;
  ssize_t r;
  idx_t lineno = 0;
  while ((r = getline(&line, &cap, f.get())) >= 0) {
    ++lineno;
    bool clean = true;
    const int n = parse_line(line, tok, MAX_NMODES + 2, &clean);
    if (n == 0 && clean) continue;          // blank / comment line
    if (!clean || n < 2 || (nmodes != 0 && n != nmodes + 1)) {
      free(line);
      throw std::runtime_error("malformed line " + std::to_string(lineno) +
                               " in " + path);
    }
    if (nmodes == 0) {
      nmodes = n - 1;
      if (nmodes < 1 || nmodes > MAX_NMODES) {
        free(line);
        throw std::runtime_error("bad mode count in " + path);
      }
    }
    for (int m = 0; m < nmodes; ++m) {
      const idx_t v = (idx_t)tok[m];
      dims[m] = std::max(dims[m], v);
      minidx = std::min(minidx, v);
    }
    ++nnz;
  }
  if (nnz == 0) {
    free(line);
    throw std::runtime_error("no nonzeros found in " + path);
  }
  const idx_t offset = (minidx == 0) ? 0 : 1;  // 0- vs 1-indexed autodetect
  for (int m = 0; m < nmodes; ++m) dims[m] += 1 - offset;

  SpTensor<V> tt(nmodes, nnz, dims);

  // pass 2: fill
  rewind(f.get());
  idx_t i = 0;
  while ((r = getline(&line, &cap, f.get())) >= 0 && i < nnz) {
    const int n = parse_line(line, tok, MAX_NMODES + 2);
    if (n < 2) continue;
    for (int m = 0; m < nmodes; ++m) tt.ind[m][i] = (idx_t)tok[m] - offset;
    tt.vals[i] = (V)tok[nmodes];
    ++i;
  }
  free(line);
  tt.nnz = i;
  return tt;
}

template <typename V>
void tns_write(const SpTensor<V> & tt, const std::string & path) {
  FilePtr f = xopen(path, "w");
  for (idx_t i = 0; i < tt.nnz; ++i) {
    for (int m = 0; m < tt.nmodes; ++m)
      fprintf(f.get(), "%llu ", (unsigned long long)(tt.ind[m][i] + 1));
    fprintf(f.get(), "%.17g\n", (double)tt.vals[i]);
  }
}

static const char BIN_MAGIC[8] = {'S','P','L','A','T','T','B','1'};

template <typename V>
void bin_write(const SpTensor<V> & tt, const std::string & path,
               int idx_bytes, int val_bytes) {
  FilePtr f = xopen(path, "wb");
  fwrite(BIN_MAGIC, 1, 8, f.get());
  const uint32_t ib = (uint32_t)idx_bytes, vb = (uint32_t)val_bytes;
  const uint64_t nm = (uint64_t)tt.nmodes, nnz = tt.nnz;
  fwrite(&ib, 4, 1, f.get());
  fwrite(&vb, 4, 1, f.get());
  fwrite(&nm, 8, 1, f.get());
  fwrite(&nnz, 8, 1, f.get());
  for (int m = 0; m < tt.nmodes; ++m) {
    const uint64_t d = tt.dims[m];
    fwrite(&d, 8, 1, f.get());
  }
  std::vector<char> buf;
  for (int m = 0; m < tt.nmodes; ++m) {
    if (idx_bytes == 8) {
      fwrite(tt.ind[m].data(), 8, nnz, f.get());
    } else {
      std::vector<uint32_t> v32(nnz);
      for (idx_t i = 0; i < nnz; ++i) v32[i] = (uint32_t)tt.ind[m][i];
      fwrite(v32.data(), 4, nnz, f.get());
    }
  }
  if ((size_t)val_bytes == sizeof(V)) {
    fwrite(tt.vals.data(), sizeof(V), nnz, f.get());
  } else if (val_bytes == 4) {
    std::vector<float> vf(nnz);
    for (idx_t i = 0; i < nnz; ++i) vf[i] = (float)tt.vals[i];
    fwrite(vf.data(), 4, nnz, f.get());
  } else {
    std::vector<double> vd(nnz);
    for (idx_t i = 0; i < nnz; ++i) vd[i] = (double)tt.vals[i];
    fwrite(vd.data(), 8, nnz, f.get());
  }
}

// Reader for the reference's .bin layout (reference src/io.h:71-88 bin_header,
// io.c:161-195 p_tt_read_binary_file): int32 magic (0 = BIN_COORD), then
// uint64 idx_width and uint64 val_width (4 or 8), then nmodes, dims[nmodes]
// and nnz at idx_width, then per-mode index arrays and the value array.
// Indices are stored 0-based (the reference writes its internal labels).
template <typename V>
SpTensor<V> bin_read_ref(FILE * fp, const std::string & path) {
  uint64_t iw = 0, vw = 0;
  if (fread(&iw, 8, 1, fp) != 1 || fread(&vw, 8, 1, fp) != 1 ||
      (iw != 4 && iw != 8) || (vw != 4 && vw != 8))
    throw std::runtime_error("bad reference binary header in " + path);
  auto rd_idx = [&](uint64_t * out, uint64_t n) {
    if (iw == 8) {
      if (fread(out, 8, n, fp) != n)
        throw std::runtime_error("truncated " + path);
    } else {
      std::vector<uint32_t> v32(n);
      if (fread(v32.data(), 4, n, fp) != n)
        throw std::runtime_error("truncated " + path);
      for (uint64_t i = 0; i < n; ++i) out[i] = v32[i];
    }
  };
  uint64_t nm = 0, nnz = 0, dims64[MAX_NMODES];
  rd_idx(&nm, 1);
  if (nm < 1 || nm > MAX_NMODES)
    throw std::runtime_error("bad nmodes in " + path);
  rd_idx(dims64, nm);
  rd_idx(&nnz, 1);
  idx_t dims[MAX_NMODES];
  for (uint64_t m = 0; m < nm; ++m) dims[m] = (idx_t)dims64[m];
  SpTensor<V> tt((int)nm, nnz, dims);
  std::vector<uint64_t> col(nnz);
  for (uint64_t m = 0; m < nm; ++m) {
    rd_idx(col.data(), nnz);
    for (uint64_t i = 0; i < nnz; ++i) tt.ind[m][i] = (idx_t)col[i];
  }
  if (vw == 4) {
    std::vector<float> vf(nnz);
    if (fread(vf.data(), 4, nnz, fp) != nnz)
      throw std::runtime_error("truncated values in " + path);
    for (idx_t i = 0; i < nnz; ++i) tt.vals[i] = (V)vf[i];
  } else {
    std::vector<double> vd(nnz);
    if (fread(vd.data(), 8, nnz, fp) != nnz)
      throw std::runtime_error("truncated values in " + path);
    for (idx_t i = 0; i < nnz; ++i) tt.vals[i] = (V)vd[i];
  }
  return tt;
}

template <typename V>
SpTensor<V> bin_read(const std::string & path) {
  FilePtr f = xopen(path, "rb");
  char magic[8];
  if (fread(magic, 1, 8, f.get()) != 8)
    throw std::runtime_error("bad binary tensor magic in " + path);
  if (memcmp(magic, BIN_MAGIC, 8) != 0) {
    // not ours: try the reference layout (int32 magic 0=COORD/1=CSF,
    // then two uint64 widths); rewind past only the int32
    int32_t rmagic;
    memcpy(&rmagic, magic, 4);
    if (rmagic == 0) {
      if (fseek(f.get(), 4, SEEK_SET) != 0)
        throw std::runtime_error("seek failed in " + path);
      return bin_read_ref<V>(f.get(), path);
    }
    if (rmagic == 1)
      throw std::runtime_error("reference BIN_CSF files are not supported; "
                               "convert from coordinate form: " + path);
    throw std::runtime_error("bad binary tensor magic in " + path);
  }
  uint32_t ib = 0, vb = 0;
  uint64_t nm = 0, nnz = 0;
  if (fread(&ib, 4, 1, f.get()) != 1 || fread(&vb, 4, 1, f.get()) != 1 ||
      fread(&nm, 8, 1, f.get()) != 1 || fread(&nnz, 8, 1, f.get()) != 1)
    throw std::runtime_error("truncated header in " + path);
  if (nm < 1 || nm > MAX_NMODES) throw std::runtime_error("bad nmodes");
  if ((ib != 4 && ib != 8) || (vb != 4 && vb != 8))
    throw std::runtime_error("bad index/value width in binary header");
  idx_t dims[MAX_NMODES];
  for (uint64_t m = 0; m < nm; ++m) {
    uint64_t d;
    if (fread(&d, 8, 1, f.get()) != 1) throw std::runtime_error("truncated dims");
    dims[m] = d;
  }
  SpTensor<V> tt((int)nm, nnz, dims);
  for (uint64_t m = 0; m < nm; ++m) {
    if (ib == 8) {
      if (fread(tt.ind[m].data(), 8, nnz, f.get()) != nnz)
        throw std::runtime_error("truncated indices");
    } else {
      std::vector<uint32_t> v32(nnz);
      if (fread(v32.data(), 4, nnz, f.get()) != nnz)
        throw std::runtime_error("truncated indices");
      for (idx_t i = 0; i < nnz; ++i) tt.ind[m][i] = v32[i];
    }
  }
  if (vb == 4) {
    std::vector<float> vf(nnz);
    if (fread(vf.data(), 4, nnz, f.get()) != nnz)
      throw std::runtime_error("truncated values");
    for (idx_t i = 0; i < nnz; ++i) tt.vals[i] = (V)vf[i];
  } else {
    std::vector<double> vd(nnz);
    if (fread(vd.data(), 8, nnz, f.get()) != nnz)
      throw std::runtime_error("truncated values");
    for (idx_t i = 0; i < nnz; ++i) tt.vals[i] = (V)vd[i];
  }
  return tt;
}

template <typename V>
SpTensor<V> tensor_load(const std::string & path) {
  const auto dot = path.rfind('.');
  const std::string ext = (dot == std::string::npos) ? "" : path.substr(dot);
  if (ext == ".bin") return bin_read<V>(path);
  return tns_read<V>(path);
}

template <typename V>
void mat_write(const V * A, idx_t nrows, int ncols, const std::string & path) {
  FilePtr f = xopen(path, "w");
  for (idx_t i = 0; i < nrows; ++i) {
    for (int j = 0; j < ncols; ++j)
      fprintf(f.get(), "%.17g%c", (double)A[i * ncols + j],
              j + 1 == ncols ? '\n' : ' ');
  }
}

template <typename V>
void vec_write(const V * v, idx_t n, const std::string & path) {
  FilePtr f = xopen(path, "w");
  for (idx_t i = 0; i < n; ++i) fprintf(f.get(), "%.17g\n", (double)v[i]);
}

#define INST(V) \
  template SpTensor<V> tns_read<V>(const std::string&); \
  template void tns_write<V>(const SpTensor<V>&, const std::string&); \
  template SpTensor<V> bin_read<V>(const std::string&); \
  template void bin_write<V>(const SpTensor<V>&, const std::string&, int, int); \
  template SpTensor<V> tensor_load<V>(const std::string&); \
  template void mat_write<V>(const V*, idx_t, int, const std::string&); \
  template void vec_write<V>(const V*, idx_t, const std::string&);
INST(float)
INST(double)
#undef INST

}  // namespace splatt
