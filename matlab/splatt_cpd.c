/* splatt_cpd: CPD-ALS over a loaded CSF handle (or a filename).
 *   K = splatt_cpd(handle_or_fname, rank[, opts])
 * K is a struct with fields U (cell of factor matrices), lambda, fit —
 * the reference's output contract (matlab/splatt_cpd.c there). opts is
 * an optional struct with fields tol, maxiters, reg, seed, verbosity.
 * On a machine with an AMD GPU the HIP engine serves the call
 * transparently (splatt_gpu_available). */
#include <string.h>

#include "mex.h"
#include "splatt.h"

static double opt_field(const mxArray * s, const char * name, double dflt) {
  const mxArray * f = mxGetField(s, 0, name);
  return f ? mxGetScalar(f) : dflt;
}

void mexFunction(int nlhs, mxArray * plhs[], int nrhs,
                 const mxArray * prhs[]) {
  if (nrhs < 2)
    mexErrMsgTxt("usage: K = splatt_cpd(csf_or_fname, rank[, opts])");
  double * opts = splatt_default_opts();
  if (nrhs >= 3 && mxIsStruct(prhs[2])) {
    opts[SPLATT_OPTION_TOLERANCE] =
        opt_field(prhs[2], "tol", opts[SPLATT_OPTION_TOLERANCE]);
    opts[SPLATT_OPTION_NITER] =
        opt_field(prhs[2], "maxiters", opts[SPLATT_OPTION_NITER]);
    opts[SPLATT_OPTION_REGULARIZE] =
        opt_field(prhs[2], "reg", opts[SPLATT_OPTION_REGULARIZE]);
    opts[SPLATT_OPTION_RANDSEED] =
        opt_field(prhs[2], "seed", opts[SPLATT_OPTION_RANDSEED]);
    opts[SPLATT_OPTION_VERBOSITY] =
        opt_field(prhs[2], "verbosity", opts[SPLATT_OPTION_VERBOSITY]);
  }
  splatt_csf * csf = NULL;
  int own = 0;
  if (mxIsChar(prhs[0])) {
    char fname[4096];
    mxGetString(prhs[0], fname, sizeof(fname));
    splatt_idx_t nm;
    if (splatt_csf_load(fname, &nm, &csf, opts) != SPLATT_SUCCESS) {
      splatt_free_opts(opts);
      mexErrMsgTxt("splatt_cpd: cannot load tensor");
    }
    own = 1;
  } else if (mxIsUint64(prhs[0])) {
    csf = (splatt_csf *)(*(unsigned long long *)mxGetData(prhs[0]));
  } else {
    splatt_free_opts(opts);
    mexErrMsgTxt("first argument must be a filename or a CSF handle");
  }
  const splatt_idx_t rank = (splatt_idx_t)mxGetScalar(prhs[1]);

  splatt_kruskal K;
  memset(&K, 0, sizeof(K));
  const int rc = splatt_cpd_als(csf, rank, opts, &K);
  if (own) splatt_free_csf(csf, opts);
  splatt_free_opts(opts);
  if (rc != SPLATT_SUCCESS) mexErrMsgTxt("splatt_cpd: factorization failed");

  const char * fields[] = {"U", "lambda", "fit"};
  plhs[0] = mxCreateStructMatrix(1, 1, 3, fields);
  mxArray * U = mxCreateCellMatrix(K.nmodes, 1);
  for (splatt_idx_t m = 0; m < K.nmodes; ++m) {
    /* row-major dims[m] x rank -> column-major MATLAB matrix */
    mxArray * A = mxCreateDoubleMatrix(K.dims[m], rank, mxREAL);
    double * dst = mxGetPr(A);
    for (splatt_idx_t i = 0; i < K.dims[m]; ++i)
      for (splatt_idx_t f = 0; f < rank; ++f)
        dst[f * K.dims[m] + i] = K.factors[m][i * rank + f];
    mxSetCell(U, m, A);
  }
  mxSetField(plhs[0], 0, "U", U);
  mxArray * lam = mxCreateDoubleMatrix(rank, 1, mxREAL);
  memcpy(mxGetPr(lam), K.lambda, sizeof(double) * rank);
  mxSetField(plhs[0], 0, "lambda", lam);
  mxSetField(plhs[0], 0, "fit", mxCreateDoubleScalar(K.fit));
  splatt_free_kruskal(&K);
}
