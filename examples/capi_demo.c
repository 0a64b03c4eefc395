/* C API walkthrough: load a tensor, build CSF, run CPD-ALS, run one
 * MTTKRP — the splatt_* surface of bin/libsplatt.so (csrc/capi/splatt.h;
 * reference include/splatt.h usage as documented in doxygen/20api.dox).
 *
 * Build & run (after `python setup.py build_ext --inplace`):
 *   gcc -O2 examples/capi_demo.c -Icsrc/capi -Lbin -lsplatt \
 *       -Wl,-rpath,$PWD/bin -o /tmp/capi_demo
 *   /tmp/capi_demo path/to/tensor.tns
 */
#include <stdio.h>
#include <stdlib.h>

#include "splatt.h"

int main(int argc, char ** argv) {
  if (argc < 2) {
    fprintf(stderr, "usage: %s TENSOR.tns [rank]\n", argv[0]);
    return 1;
  }
  const int rank = argc > 2 ? atoi(argv[2]) : 8;

  double * opts = splatt_default_opts();
  opts[SPLATT_OPTION_NITER] = 20;
  opts[SPLATT_OPTION_TOLERANCE] = 1e-5;

  splatt_idx_t nmodes = 0;
  splatt_csf * csf = NULL;
  if (splatt_csf_load(argv[1], &nmodes, &csf, opts) != SPLATT_SUCCESS) {
    fprintf(stderr, "load failed: %s\n", argv[1]);
    return 1;
  }
  printf("loaded %s: %d modes\n", argv[1], (int)nmodes);

  splatt_kruskal factored;
  if (splatt_cpd_als(csf, rank, opts, &factored) != SPLATT_SUCCESS) {
    fprintf(stderr, "cpd failed\n");
    return 1;
  }
  printf("rank-%d CPD fit: %.5f\n", rank, factored.fit);

  /* one standalone MTTKRP against the factor matrices, mode 0 */
  splatt_idx_t maxdim = 0;
  for (splatt_idx_t m = 0; m < nmodes; ++m)
    if (factored.dims[m] > maxdim) maxdim = factored.dims[m];
  double * out = (double *)malloc(maxdim * rank * sizeof(double));
  splatt_mttkrp_ws * ws = splatt_mttkrp_alloc_ws(csf, rank, opts);
  (void)ws;  /* flat engine needs no scratch; kept for API parity */
  if (splatt_mttkrp(0, rank, csf, factored.factors, out, opts)
      != SPLATT_SUCCESS) {
    fprintf(stderr, "mttkrp failed\n");
    return 1;
  }
  printf("mttkrp mode 0: out[0][0] = %g\n", out[0]);

  splatt_mttkrp_free_ws(ws);
  free(out);
  splatt_free_kruskal(&factored);
  splatt_free_csf(csf, opts);
  splatt_free_opts(opts);
  return 0;
}
