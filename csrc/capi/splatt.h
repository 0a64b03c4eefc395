/**
 * splatt.h — public C API of the MI355X-native sparse tensor
 * factorization engine.
 *
 * Capability parity with SPLATT's public API (reference
 * include/splatt.h + include/splatt/api_*.h): the same entry points —
 * splatt_default_opts / splatt_free_opts, splatt_csf_load /
 * splatt_csf_convert / splatt_free_csf, splatt_cpd_als /
 * splatt_free_kruskal, splatt_mttkrp, splatt_version_* — with the same
 * options-array calling convention (double[SPLATT_OPTION_NOPTIONS]) and
 * the same Kruskal output contract. The CSF handle is opaque here (the
 * rebuild's CSF is a flat per-level array set designed for GPU
 * residency, not the reference's tile struct tree).
 */
#ifndef SPLATT_AMD_SPLATT_H
#define SPLATT_AMD_SPLATT_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef uint64_t splatt_idx_t;
typedef double splatt_val_t;

#define SPLATT_MAX_NMODES 8

/* Numeric values below match the reference public header exactly
 * (reference include/splatt/types_config.h:103-215) so clients compiled
 * against either header agree on option slots and return codes:
 * options NTHREADS=0, TOLERANCE=1, REGULARIZE=2, NITER=3, VERBOSITY=4,
 * then RANDSEED..COMM; SPLATT_SUCCESS=1 with positive error codes. */
typedef enum splatt_error_type {
  SPLATT_SUCCESS = 1,
  SPLATT_ERROR_BADINPUT = 2,
  SPLATT_ERROR_NOMEMORY = 3,
} splatt_error_type;

typedef enum splatt_verbosity_type {
  SPLATT_VERBOSITY_NONE = 0,
  SPLATT_VERBOSITY_LOW = 1,
  SPLATT_VERBOSITY_HIGH = 2,
  SPLATT_VERBOSITY_MAX = 3,
} splatt_verbosity_type;

typedef enum splatt_csf_type {
  SPLATT_CSF_ONEMODE = 0,
  SPLATT_CSF_TWOMODE = 1,
  SPLATT_CSF_ALLMODE = 2,
} splatt_csf_type;

typedef enum splatt_tile_type {
  SPLATT_NOTILE = 0,
  SPLATT_DENSETILE = 1,
  SPLATT_SYNCTILE = 2,   /* deprecated in the reference; accepted for
                            compatibility (the engine chooses tiling) */
  SPLATT_COOPTILE = 3,
} splatt_tile_type;

typedef enum splatt_decomp_type {
  SPLATT_DECOMP_COARSE = 0,
  SPLATT_DECOMP_MEDIUM = 1,
  SPLATT_DECOMP_FINE = 2,
} splatt_decomp_type;

typedef enum splatt_comm_type {
  SPLATT_COMM_POINT2POINT = 0,
  SPLATT_COMM_ALL2ALL = 1,
} splatt_comm_type;

typedef enum splatt_option_type {
  SPLATT_OPTION_NTHREADS = 0,   /* worker threads (0 = hw default)    */
  SPLATT_OPTION_TOLERANCE,      /* convergence tolerance (1e-5)       */
  SPLATT_OPTION_REGULARIZE,     /* ridge term on the Gram diagonal (0)*/
  SPLATT_OPTION_NITER,          /* max ALS iterations (50)            */
  SPLATT_OPTION_VERBOSITY,      /* splatt_verbosity_type (LOW)        */
  SPLATT_OPTION_RANDSEED,       /* RNG seed (fixed default)           */
  SPLATT_OPTION_CSF_ALLOC,      /* splatt_csf_type (TWOMODE)          */
  SPLATT_OPTION_TILE,           /* splatt_tile_type (NOTILE)            */
  SPLATT_OPTION_TILELEVEL,      /* advisory (GPU build picks LDS tiles) */
  SPLATT_OPTION_PRIVTHRESH,     /* advisory (GPU folds runs in regs)    */
  SPLATT_OPTION_DECOMP,         /* splatt_decomp_type (MEDIUM)          */
  SPLATT_OPTION_COMM,           /* splatt_comm_type (ALL2ALL analog)    */
  SPLATT_OPTION_NOPTIONS
} splatt_option_type;

/** Opaque CSF tensor handle. Calls taking the same handle are not
 * thread-safe against each other (the handle lazily caches
 * device-resident streams on first GPU use); distinct handles are
 * independent. */
typedef struct splatt_csf splatt_csf;

/** Kruskal tensor: the CPD output (parity: reference structs.h:25-44). */
typedef struct splatt_kruskal {
  splatt_idx_t rank;
  splatt_val_t * factors[SPLATT_MAX_NMODES]; /* row-major dims[m] x rank */
  splatt_val_t * lambda;
  splatt_idx_t nmodes;
  splatt_idx_t dims[SPLATT_MAX_NMODES];
  double fit;
} splatt_kruskal;

/* ------------------------------------------------------------ options */
double * splatt_default_opts(void);
void splatt_free_opts(double * opts);

/* ---------------------------------------------------------------- csf */
/** Load a tensor file (.tns/.coo/.bin) into CSF form. `*tensors` receives
 * an array of 1, 2 or nmodes CSF structures per SPLATT_OPTION_CSF_ALLOC. */
int splatt_csf_load(const char * fname, splatt_idx_t * nmodes,
                    splatt_csf ** tensors, const double * options);

/** Convert an in-memory coordinate tensor to CSF. */
int splatt_csf_convert(splatt_idx_t nmodes, splatt_idx_t nnz,
                       splatt_idx_t ** inds, splatt_val_t * vals,
                       splatt_csf ** tensors, const double * options);

void splatt_free_csf(splatt_csf * tensors, const double * options);

/* number of modes / dims accessors for the opaque handle */
splatt_idx_t splatt_csf_nmodes(const splatt_csf * csf);
splatt_idx_t splatt_csf_nnz(const splatt_csf * csf);
void splatt_csf_dims(const splatt_csf * csf, splatt_idx_t * dims);

/* -------------------------------------------------------- factorization */
int splatt_cpd_als(const splatt_csf * tensors, splatt_idx_t nfactors,
                   const double * options, splatt_kruskal * factored);
void splatt_free_kruskal(splatt_kruskal * factored);

/* ------------------------------------------------------------- kernels */
/** One MTTKRP: mats[m] row-major dims[m] x ncolumns; result for `mode`
 * written into matout (parity: api_kernels.h splatt_mttkrp). */
int splatt_mttkrp(splatt_idx_t mode, splatt_idx_t ncolumns,
                  const splatt_csf * tensors, splatt_val_t ** matrices,
                  splatt_val_t * matout, const double * options);

/** MTTKRP workspace (parity: api_kernels.h splatt_mttkrp_alloc_ws). The
 * flat-array engine needs no scratch, so this is a thin mode->CSF
 * dispatch record kept for API compatibility. */
typedef struct splatt_mttkrp_ws {
  splatt_idx_t num_csf;
  splatt_idx_t mode_csf_map[SPLATT_MAX_NMODES];
} splatt_mttkrp_ws;

splatt_mttkrp_ws * splatt_mttkrp_alloc_ws(const splatt_csf * tensors,
                                          splatt_idx_t ncolumns,
                                          const double * options);
void splatt_mttkrp_free_ws(splatt_mttkrp_ws * ws);

/* ------------------------------------------------------------- version */
/** 1 when an AMD GPU is visible and the HIP engine will serve
 * splatt_cpd_als (override with SPLATT_CAPI_CPU=1). */
int splatt_gpu_available(void);

int splatt_version_major(void);
int splatt_version_minor(void);
int splatt_version_subminor(void);

#ifdef __cplusplus
}
#endif

#endif
