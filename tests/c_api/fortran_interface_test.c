/* Exercise the Fortran (trailing-underscore, pass-by-reference) entry
 * points of libquda_amd_c.so from C — the calling convention an
 * implicit-interface Fortran compiler emits (role of the reference's
 * quda_fortran.F90 wrappers; no gfortran in this image, so the ABI is
 * tested at the symbol level). */
#include <math.h>
#include <stdio.h>
#include <stdlib.h>

#include "../../include/quda_amd.h"

void init_quda_(int *device);
void end_quda_(void);
void new_quda_gauge_param_(QudaAmdGaugeParam *p);
void new_quda_invert_param_(QudaAmdInvertParam *p);
void load_gauge_quda_(void *h_gauge, QudaAmdGaugeParam *param, int *ierr);
void invert_quda_(void *h_x, void *h_b, QudaAmdInvertParam *param, int *ierr);
void plaq_quda_(double *plaq, int *ierr);

#define L 4
#define VCB (L * L * L * L / 2)
typedef struct { double re, im; } cplx;

int main(void) {
  int dev = 0, ierr = 0;
  size_t glinks = (size_t)4 * 2 * VCB;
  cplx *gauge = (cplx *)calloc(glinks * 9, sizeof(cplx));
  for (size_t l = 0; l < glinks; ++l)
    for (int c = 0; c < 3; ++c) gauge[l * 9 + c * 3 + c].re = 1.0;

  init_quda_(&dev);
  QudaAmdGaugeParam gp;
  new_quda_gauge_param_(&gp);
  gp.X[0] = gp.X[1] = gp.X[2] = gp.X[3] = L;
  gp.cuda_prec_sloppy = QUDA_AMD_DOUBLE;
  gp.reconstruct_sloppy = QUDA_AMD_RECON_NO;
  load_gauge_quda_(gauge, &gp, &ierr);
  if (ierr) { fprintf(stderr, "load_gauge_quda_ failed\n"); return 1; }

  double plaq[3];
  plaq_quda_(plaq, &ierr);
  if (ierr || fabs(plaq[0] - 1.0) > 1e-12) return 2;

  QudaAmdInvertParam ip;
  new_quda_invert_param_(&ip);
  ip.kappa = 0.1;
  ip.inv_type = QUDA_AMD_CGNR_INVERTER;
  ip.tol = 1e-8;
  ip.cuda_prec_sloppy = QUDA_AMD_DOUBLE;
  size_t ns = (size_t)2 * VCB * 12;
  cplx *b = (cplx *)calloc(ns, sizeof(cplx));
  cplx *x = (cplx *)calloc(ns, sizeof(cplx));
  for (size_t i = 0; i < ns; ++i) b[i].re = (double)((i * 2654435761u) % 97) / 97.0 - 0.5;
  invert_quda_(x, b, &ip, &ierr);
  if (ierr || ip.true_res > 1e-6 || ip.iter <= 0) return 3;

  /* round-2 Fortran surface: observables + smearing + HMC + eigensolve */
  double qtop, energy[2];
  gauge_observables_quda_(plaq, &qtop, energy, &ierr);
  if (ierr || fabs(plaq[0] - 1.0) > 1e-12) return 4;
  int stype = 1 /* stout */, nstep = 1;
  double coeff = 0.05;
  perform_gauge_smear_quda_(&stype, &nstep, &coeff, &ierr);
  if (ierr) return 5;
  project_su3_quda_(&ierr);
  if (ierr) return 6;
  cplx *mom = (cplx *)calloc((size_t)4 * 2 * VCB * 9, sizeof(cplx));
  long seed = 12;
  gauss_mom_quda_(mom, &seed, &ierr);
  if (ierr) return 7;
  double mact = 0;
  mom_action_quda_(&mact, mom, &ierr);
  if (ierr || mact <= 0) return 8;
  QudaAmdEigParam ep;
  new_quda_eig_param_(&ep);
  ep.n_ev = 2;
  ep.n_kr = 8;
  ep.tol = 1e-4;
  ip.solution_type = QUDA_AMD_MATPC_SOLUTION;
  double er[2], ei[2];
  eigensolve_quda_(er, ei, &ip, &ep, &ierr);
  if (ierr || er[0] <= 0) return 9;
  free(mom);

  printf("fortran_interface_test: solve %d iters res %.2e — ALL PASSED\n",
         ip.iter, ip.true_res);
  end_quda_();
  return 0;
}
