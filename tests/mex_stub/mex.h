/* Minimal MEX API stub: lets CI compile-check the matlab sources without an
 * Octave/MATLAB toolchain (tests/test_capi.py::test_mex_sources_compile).
 * Declarations only mirror the subset the bindings use. */
#ifndef SPLATT_TEST_MEX_STUB_H
#define SPLATT_TEST_MEX_STUB_H
#include <stddef.h>

typedef struct mxArray_tag mxArray;
typedef enum { mxREAL = 0, mxCOMPLEX } mxComplexity;
typedef enum { mxUINT64_CLASS = 1 } mxClassID;

int mxIsChar(const mxArray *);
int mxIsStruct(const mxArray *);
int mxIsCell(const mxArray *);
int mxIsUint64(const mxArray *);
int mxGetString(const mxArray *, char *, size_t);
double mxGetScalar(const mxArray *);
void * mxGetData(const mxArray *);
double * mxGetPr(const mxArray *);
size_t mxGetM(const mxArray *);
size_t mxGetN(const mxArray *);
size_t mxGetNumberOfElements(const mxArray *);
mxArray * mxGetField(const mxArray *, size_t, const char *);
mxArray * mxGetCell(const mxArray *, size_t);
void mxSetCell(mxArray *, size_t, mxArray *);
void mxSetField(mxArray *, size_t, const char *, mxArray *);
mxArray * mxCreateNumericMatrix(size_t, size_t, mxClassID, mxComplexity);
mxArray * mxCreateDoubleMatrix(size_t, size_t, mxComplexity);
mxArray * mxCreateDoubleScalar(double);
mxArray * mxCreateCellMatrix(size_t, size_t);
mxArray * mxCreateStructMatrix(size_t, size_t, int, const char **);
void mexErrMsgTxt(const char *);
#endif
