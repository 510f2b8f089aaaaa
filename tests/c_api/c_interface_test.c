/* C-linkage end-to-end test of the quda_amd C ABI (role of the
 * reference's tests/c_interface_test.cpp): pure C99, links only against
 * libquda_amd_c.so.
 *
 * Builds a free (unit-link) gauge field on an 8^4 lattice, loads it,
 * checks plaqQuda() == 1, computes the clover term, solves the
 * Wilson-clover system M x = b with mixed-precision CG to 1e-8, applies
 * MatQuda to the solution and verifies ||M x - b|| / ||b|| < 1e-6, then
 * runs a MATPC solve and a 3-shift multishift solve.
 *
 * Exit code 0 = all checks passed.
 */
#include <math.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>

#include "../../include/quda_amd.h"

#define L 8
#define VCB (L * L * L * L / 2)
#define CHECK(cond, msg)                                                     \
  do {                                                                       \
    if (!(cond)) {                                                           \
      fprintf(stderr, "FAIL: %s (%s)\n", msg, qudaAmdLastError());           \
      return 1;                                                              \
    }                                                                        \
  } while (0)

typedef struct { double re, im; } cplx;

/* deterministic pseudo-random source (LCG; no libc rand state issues) */
static unsigned long long lcg_state = 12345;
static double lcg(void) {
  lcg_state = lcg_state * 6364136223846793005ULL + 1442695040888963407ULL;
  return ((double)(lcg_state >> 11) / 9007199254740992.0) * 2.0 - 1.0;
}

int main(void) {
  /* ---- unit gauge field: [4][2][VCB][3][3] ---- */
  size_t glinks = (size_t)4 * 2 * VCB;
  cplx *gauge = (cplx *)calloc(glinks * 9, sizeof(cplx));
  for (size_t l = 0; l < glinks; ++l)
    for (int c = 0; c < 3; ++c) gauge[l * 9 + c * 3 + c].re = 1.0;

  CHECK(initQuda(0) == 0, "initQuda");

  QudaAmdGaugeParam gp = newQudaAmdGaugeParam();
  gp.X[0] = gp.X[1] = gp.X[2] = gp.X[3] = L;
  gp.cuda_prec_sloppy = QUDA_AMD_DOUBLE; /* free field: exact checks */
  gp.reconstruct_sloppy = QUDA_AMD_RECON_NO;
  CHECK(loadGaugeQuda(gauge, &gp) == 0, "loadGaugeQuda");

  double plaq[3];
  CHECK(plaqQuda(plaq) == 0, "plaqQuda");
  CHECK(fabs(plaq[0] - 1.0) < 1e-12, "unit-field plaquette == 1");

  /* ---- Wilson-clover CG solve on a random source ---- */
  QudaAmdInvertParam ip = newQudaAmdInvertParam();
  ip.dslash_type = QUDA_AMD_CLOVER_WILSON_DSLASH;
  ip.inv_type = QUDA_AMD_CGNR_INVERTER;
  ip.kappa = 0.11;
  ip.clover_csw = 1.0;
  ip.tol = 1e-9;
  ip.maxiter = 500;
  ip.cuda_prec_sloppy = QUDA_AMD_DOUBLE;
  CHECK(loadCloverQuda(NULL, NULL, &ip) == 0, "loadCloverQuda(compute)");

  size_t ns = (size_t)2 * VCB * 12;
  cplx *b = (cplx *)malloc(ns * sizeof(cplx));
  cplx *x = (cplx *)malloc(ns * sizeof(cplx));
  cplx *mx = (cplx *)malloc(ns * sizeof(cplx));
  for (size_t i = 0; i < ns; ++i) { b[i].re = lcg(); b[i].im = lcg(); }

  CHECK(invertQuda(x, b, &ip) == 0, "invertQuda");
  CHECK(ip.iter > 0, "iteration count recorded");
  CHECK(ip.true_res < 1e-7, "reported true residual under tolerance");

  /* independent check through MatQuda */
  CHECK(MatQuda(mx, x, &ip) == 0, "MatQuda");
  double r2 = 0, b2 = 0;
  for (size_t i = 0; i < ns; ++i) {
    double dr = mx[i].re - b[i].re, di = mx[i].im - b[i].im;
    r2 += dr * dr + di * di;
    b2 += b[i].re * b[i].re + b[i].im * b[i].im;
  }
  CHECK(sqrt(r2 / b2) < 1e-6, "||Mx-b||/||b|| < 1e-6");
  printf("wilson-clover solve: %d iters, true_res %.2e, check %.2e\n",
         ip.iter, ip.true_res, sqrt(r2 / b2));

  /* ---- even-odd preconditioned solve path ---- */
  ip.solution_type = QUDA_AMD_MATPC_SOLUTION;
  ip.inv_type = QUDA_AMD_CG_INVERTER;
  CHECK(invertQuda(x, b, &ip) == 0, "invertQuda(MATPC)");
  ip.solution_type = QUDA_AMD_MAT_SOLUTION; /* check the FULL operator */
  CHECK(MatQuda(mx, x, &ip) == 0, "MatQuda(pc check)");
  r2 = 0;
  for (size_t i = 0; i < ns; ++i) {
    double dr = mx[i].re - b[i].re, di = mx[i].im - b[i].im;
    r2 += dr * dr + di * di;
  }
  CHECK(sqrt(r2 / b2) < 1e-6, "MATPC ||Mx-b||/||b|| < 1e-6");
  printf("matpc solve: %d iters, check %.2e\n", ip.iter, sqrt(r2 / b2));

  /* ---- multishift (single-parity source) ---- */
  ip.solution_type = QUDA_AMD_MATPC_SOLUTION;
  ip.dslash_type = QUDA_AMD_WILSON_DSLASH;
  double shifts[3] = {0.0, 0.1, 0.5};
  size_t nsp = (size_t)VCB * 12;
  cplx *xs0 = (cplx *)malloc(nsp * sizeof(cplx));
  cplx *xs1 = (cplx *)malloc(nsp * sizeof(cplx));
  cplx *xs2 = (cplx *)malloc(nsp * sizeof(cplx));
  void *xs[3] = {xs0, xs1, xs2};
  CHECK(invertMultiShiftQuda(xs, b, &ip, shifts, 3) == 0,
        "invertMultiShiftQuda");
  CHECK(ip.iter > 0, "multishift iterated");
  /* shifted solutions must differ */
  double d01 = 0;
  for (size_t i = 0; i < nsp; ++i) {
    double dr = xs0[i].re - xs1[i].re;
    d01 += dr * dr;
  }
  CHECK(d01 > 1e-12, "shifted solutions differ");
  printf("multishift: %d iters\n", ip.iter);

  /* ---- kappa-convention check: standard quda.h normalization means
   *      M (constant spinor) = (1 - 8 kappa) (constant spinor) on the
   *      free field ---- */
  ip.dslash_type = QUDA_AMD_WILSON_DSLASH;
  ip.solution_type = QUDA_AMD_MAT_SOLUTION;
  ip.kappa = 0.1;
  for (size_t i = 0; i < ns; ++i) { b[i].re = 1.0; b[i].im = 0.0; }
  CHECK(MatQuda(mx, b, &ip) == 0, "MatQuda(convention)");
  CHECK(fabs(mx[0].re - (1.0 - 8.0 * 0.1)) < 1e-10,
        "standard Wilson kappa normalization (M psi0 = (1-8k) psi0)");
  for (size_t i = 0; i < ns; ++i) { b[i].re = lcg(); b[i].im = lcg(); }

  /* ---- dslashQuda parity application on the free field ---- */
  ip.solution_type = QUDA_AMD_MAT_SOLUTION;
  CHECK(dslashQuda(mx, b, &ip, 0) == 0, "dslashQuda");

  CHECK(saveGaugeQuda(gauge, &gp) == 0, "saveGaugeQuda");
  for (int c = 0; c < 3; ++c)
    CHECK(fabs(gauge[c * 3 + c].re - 1.0) < 1e-12, "saved gauge round-trip");

  /* ---- round-2 surface: observables, smearing, HMC, eigensolve, MG,
   *      gauge fixing, contraction ---- */
  double qcharge, energy[2];
  CHECK(gaugeObservablesQuda(plaq, &qcharge, energy) == 0,
        "gaugeObservablesQuda");
  CHECK(fabs(plaq[0] - 1.0) < 1e-12 && fabs(qcharge) < 1e-8,
        "unit-field observables");

  CHECK(performGaugeSmearQuda(QUDA_AMD_SMEAR_STOUT, 2, 0.1) == 0,
        "stout smear");
  CHECK(performGaugeSmearQuda(QUDA_AMD_SMEAR_WILSON_FLOW, 2, 0.02) == 0,
        "wilson flow");
  CHECK(projectSU3Quda() == 0, "projectSU3Quda");
  CHECK(plaqQuda(plaq) == 0 && fabs(plaq[0] - 1.0) < 1e-10,
        "unit field invariant under smearing");

  cplx *mom = (cplx *)malloc(glinks * 9 * sizeof(cplx));
  CHECK(gaussMomQuda(mom, 77) == 0, "gaussMomQuda");
  double mact = 0;
  CHECK(momActionQuda(&mact, mom) == 0 && mact > 0, "momActionQuda");
  CHECK(momResidentQuda(mom) == 0, "momResidentQuda(set)");
  CHECK(momResidentQuda(NULL) == 0, "momResidentQuda(clear)");
  cplx *force = (cplx *)malloc(glinks * 9 * sizeof(cplx));
  CHECK(computeGaugeForceQuda(force, 5.5) == 0, "computeGaugeForceQuda");
  double fmax = 0;
  for (size_t i = 0; i < glinks * 9; ++i) {
    if (fabs(force[i].re) > fmax) fmax = fabs(force[i].re);
    if (fabs(force[i].im) > fmax) fmax = fabs(force[i].im);
  }
  CHECK(fmax < 1e-10, "unit-field gauge force vanishes");
  CHECK(updateGaugeFieldQuda(mom, 0.0) == 0, "updateGaugeFieldQuda(dt=0)");
  CHECK(plaqQuda(plaq) == 0 && fabs(plaq[0] - 1.0) < 1e-12,
        "dt=0 update is the identity");

  QudaAmdEigParam ep = newQudaAmdEigParam();
  ep.n_ev = 4;
  ep.n_kr = 16;
  ep.tol = 1e-6;
  ip.dslash_type = QUDA_AMD_WILSON_DSLASH;
  ip.solution_type = QUDA_AMD_MATPC_SOLUTION;
  ip.kappa = 0.10;
  double ev_re[4], ev_im[4];
  CHECK(eigensolveQuda(ev_re, ev_im, NULL, &ip, &ep) == 0,
        "eigensolveQuda");
  CHECK(ev_re[0] > 0 && ev_re[0] <= ev_re[1] + 1e-12,
        "MdagM spectrum positive and sorted");
  printf("eigensolve: lambda_min = %.6f\n", ev_re[0]);

  QudaAmdMultigridParam mp = newQudaAmdMultigridParam();
  mp.n_vec = 2;
  ip.solution_type = QUDA_AMD_MAT_SOLUTION;
  ip.inv_type = QUDA_AMD_GCR_INVERTER;
  ip.kappa = 0.11;
  ip.tol = 1e-8;
  ip.maxiter = 200;
  void *mg = newMultigridQuda(&ip, &mp);
  CHECK(mg != NULL, "newMultigridQuda");
  ip.preconditioner = mg;
  CHECK(invertQuda(x, b, &ip) == 0, "invertQuda(MG-preconditioned)");
  CHECK(ip.iter > 0, "MG solve iterated");
  ip.preconditioner = NULL;
  CHECK(MatQuda(mx, x, &ip) == 0, "MatQuda(mg check)");
  r2 = 0;
  for (size_t i = 0; i < ns; ++i) {
    double dr = mx[i].re - b[i].re, di = mx[i].im - b[i].im;
    r2 += dr * dr + di * di;
  }
  CHECK(sqrt(r2 / b2) < 1e-6, "MG ||Mx-b||/||b|| < 1e-6");
  printf("mg-gcr solve: %d iters, check %.2e\n", ip.iter, sqrt(r2 / b2));
  CHECK(destroyMultigridQuda(mg) == 0, "destroyMultigridQuda");

  CHECK(computeGaugeFixingOVRQuda(4, 3, 1e-10) == 0,
        "computeGaugeFixingOVRQuda");

  /* contraction of two constant propagator fields: C[x,s,s'] = 3 */
  for (size_t i = 0; i < ns; ++i) { b[i].re = 1.0; b[i].im = 0.0; }
  cplx *corr = (cplx *)malloc((size_t)L * L * L * L * 16 * sizeof(cplx));
  CHECK(contractQuda(corr, b, b, &ip, 0) == 0, "contractQuda");
  CHECK(fabs(corr[0].re - 3.0) < 1e-12, "open-spin contraction value");
  free(corr);
  free(mom);
  free(force);

  CHECK(freeGaugeQuda() == 0, "freeGaugeQuda");
  CHECK(endQuda() == 0, "endQuda");
  printf("c_interface_test: ALL PASSED\n");
  return 0;
}
