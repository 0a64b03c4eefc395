/* splatt_load: load a .tns/.bin tensor into CSF and return an opaque
 * handle (uint64). MEX shim over the C API (csrc/capi/splatt.h) —
 * capability parity with the reference's matlab/splatt_load.c. Build
 * with matlab/make_splatt.m (needs an Octave/MATLAB toolchain; none
 * ships in this repo's CI image, so these bindings are provided
 * untested there). */
#include <string.h>

#include "mex.h"
#include "splatt.h"

void mexFunction(int nlhs, mxArray * plhs[], int nrhs,
                 const mxArray * prhs[]) {
  if (nrhs < 1 || !mxIsChar(prhs[0]))
    mexErrMsgTxt("usage: handle = splatt_load('tensor.tns')");
  char fname[4096];
  mxGetString(prhs[0], fname, sizeof(fname));
  double * opts = splatt_default_opts();
  splatt_idx_t nmodes;
  splatt_csf * csf = NULL;
  if (splatt_csf_load(fname, &nmodes, &csf, opts) != SPLATT_SUCCESS) {
    splatt_free_opts(opts);
    mexErrMsgTxt("splatt_load: cannot load tensor");
  }
  splatt_free_opts(opts);
  plhs[0] = mxCreateNumericMatrix(1, 1, mxUINT64_CLASS, mxREAL);
  *(unsigned long long *)mxGetData(plhs[0]) = (unsigned long long)csf;
  if (nlhs > 1) {
    plhs[1] = mxCreateDoubleScalar((double)nmodes);
  }
}
