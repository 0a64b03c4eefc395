/* splatt_mttkrp: one MTTKRP against a loaded CSF handle.
 *   M = splatt_mttkrp(handle, mats_cell, mode)   (mode is 1-indexed)
 * mats_cell{m} is dims[m] x rank (column-major; transposed into the
 * engine's row-major layout and back). */
#include <stdlib.h>
#include <string.h>

#include "mex.h"
#include "splatt.h"

void mexFunction(int nlhs, mxArray * plhs[], int nrhs,
                 const mxArray * prhs[]) {
  if (nrhs < 3 || !mxIsUint64(prhs[0]) || !mxIsCell(prhs[1]))
    mexErrMsgTxt("usage: M = splatt_mttkrp(handle, {U1..Un}, mode)");
  splatt_csf * csf =
      (splatt_csf *)(*(unsigned long long *)mxGetData(prhs[0]));
  const splatt_idx_t nmodes = splatt_csf_nmodes(csf);
  splatt_idx_t dims[SPLATT_MAX_NMODES];
  splatt_csf_dims(csf, dims);
  const splatt_idx_t mode = (splatt_idx_t)mxGetScalar(prhs[2]) - 1;
  if (mode >= nmodes) mexErrMsgTxt("mode out of range");
  if ((splatt_idx_t)mxGetNumberOfElements(prhs[1]) != nmodes)
    mexErrMsgTxt("need one factor matrix per mode");

  splatt_idx_t rank = 0;
  splatt_val_t * mats[SPLATT_MAX_NMODES] = {0};
  for (splatt_idx_t m = 0; m < nmodes; ++m) {
    const mxArray * A = mxGetCell(prhs[1], m);
    if (!A || mxGetM(A) != dims[m])
      mexErrMsgTxt("factor matrix has wrong row count");
    if (m == 0) rank = (splatt_idx_t)mxGetN(A);
    const double * src = mxGetPr(A);
    mats[m] = (splatt_val_t *)malloc(sizeof(double) * dims[m] * rank);
    for (splatt_idx_t i = 0; i < dims[m]; ++i)
      for (splatt_idx_t f = 0; f < rank; ++f)
        mats[m][i * rank + f] = src[f * dims[m] + i];
  }
  splatt_val_t * out =
      (splatt_val_t *)calloc(dims[mode] * rank, sizeof(double));
  double * opts = splatt_default_opts();
  const int rc = splatt_mttkrp(mode, rank, csf, mats, out, opts);
  splatt_free_opts(opts);
  for (splatt_idx_t m = 0; m < nmodes; ++m) free(mats[m]);
  if (rc != SPLATT_SUCCESS) {
    free(out);
    mexErrMsgTxt("splatt_mttkrp failed");
  }
  plhs[0] = mxCreateDoubleMatrix(dims[mode], rank, mxREAL);
  double * dst = mxGetPr(plhs[0]);
  for (splatt_idx_t i = 0; i < dims[mode]; ++i)
    for (splatt_idx_t f = 0; f < rank; ++f)
      dst[f * dims[mode] + i] = out[i * rank + f];
  free(out);
}
