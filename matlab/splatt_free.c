/* splatt_free: release a CSF handle from splatt_load. */
#include "mex.h"
#include "splatt.h"

void mexFunction(int nlhs, mxArray * plhs[], int nrhs,
                 const mxArray * prhs[]) {
  (void)nlhs; (void)plhs;
  if (nrhs < 1 || !mxIsUint64(prhs[0]))
    mexErrMsgTxt("usage: splatt_free(handle)");
  splatt_csf * csf =
      (splatt_csf *)(*(unsigned long long *)mxGetData(prhs[0]));
  splatt_free_csf(csf, NULL);
}
