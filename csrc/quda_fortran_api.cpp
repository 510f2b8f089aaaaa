// Fortran bindings for the quda_amd C ABI (role of the reference's
// lib/quda_fortran.F90 + lib/interface/fortran_interface.cpp): classic
// trailing-underscore, pass-by-reference entry points callable from
// FORTRAN 77 / implicit-interface Fortran, plus the iso_c_binding module
// in fortran/quda_amd.f90 for modern code. Compiled into
// libquda_amd_c.so next to the C symbols it wraps.
#include "../include/quda_amd.h"

extern "C" {

void init_quda_(int *device) { initQuda(*device); }
void end_quda_(void) { endQuda(); }
void free_gauge_quda_(void) { freeGaugeQuda(); }

void new_quda_gauge_param_(QudaAmdGaugeParam *p) { *p = newQudaAmdGaugeParam(); }
void new_quda_invert_param_(QudaAmdInvertParam *p) { *p = newQudaAmdInvertParam(); }

void load_gauge_quda_(void *h_gauge, QudaAmdGaugeParam *param, int *ierr) {
  *ierr = loadGaugeQuda(h_gauge, param);
}

void save_gauge_quda_(void *h_gauge, QudaAmdGaugeParam *param, int *ierr) {
  *ierr = saveGaugeQuda(h_gauge, param);
}

void load_clover_quda_(void *h_clover, void *h_clovinv,
                       QudaAmdInvertParam *param, int *ierr) {
  // Fortran side passes 0 (NULL by value convention differs): treat a
  // null pointer OR an all-null reference as "compute from gauge"
  *ierr = loadCloverQuda(h_clover, h_clovinv, param);
}

void invert_quda_(void *h_x, void *h_b, QudaAmdInvertParam *param,
                  int *ierr) {
  *ierr = invertQuda(h_x, h_b, param);
}

void dslash_quda_(void *h_out, void *h_in, QudaAmdInvertParam *param,
                  int *parity, int *ierr) {
  *ierr = dslashQuda(h_out, h_in, param, *parity);
}

void mat_quda_(void *h_out, void *h_in, QudaAmdInvertParam *param,
               int *ierr) {
  *ierr = MatQuda(h_out, h_in, param);
}

void mat_dag_mat_quda_(void *h_out, void *h_in, QudaAmdInvertParam *param,
                       int *ierr) {
  *ierr = MatDagMatQuda(h_out, h_in, param);
}

void plaq_quda_(double *plaq, int *ierr) { *ierr = plaqQuda(plaq); }

// ---- round-2 surface (eigensolve, MG, HMC, smearing, observables,
//      gauge fixing, contractions) ----

void new_quda_eig_param_(QudaAmdEigParam *p) { *p = newQudaAmdEigParam(); }
void new_quda_multigrid_param_(QudaAmdMultigridParam *p) {
  *p = newQudaAmdMultigridParam();
}

void eigensolve_quda_(double *evals_re, double *evals_im,
                      QudaAmdInvertParam *ip, QudaAmdEigParam *ep,
                      int *ierr) {
  *ierr = eigensolveQuda(evals_re, evals_im, nullptr, ip, ep);
}

void new_multigrid_quda_(void **handle, QudaAmdInvertParam *ip,
                         QudaAmdMultigridParam *mp, int *ierr) {
  *handle = newMultigridQuda(ip, mp);
  *ierr = *handle ? 0 : -1;
}

void update_multigrid_quda_(void **handle, QudaAmdInvertParam *ip,
                            int *ierr) {
  *ierr = updateMultigridQuda(*handle, ip);
}

void destroy_multigrid_quda_(void **handle, int *ierr) {
  *ierr = destroyMultigridQuda(*handle);
}

void compute_gauge_force_quda_(void *h_mom, double *beta, int *ierr) {
  *ierr = computeGaugeForceQuda(h_mom, *beta);
}

void update_gauge_field_quda_(void *h_mom, double *dt, int *ierr) {
  *ierr = updateGaugeFieldQuda(h_mom, *dt);
}

void mom_action_quda_(double *action, void *h_mom, int *ierr) {
  *ierr = momActionQuda(action, h_mom);
}

void mom_resident_quda_(void *h_mom, int *ierr) {
  *ierr = momResidentQuda(h_mom);
}

void gauss_mom_quda_(void *h_mom, long *seed, int *ierr) {
  *ierr = gaussMomQuda(h_mom, *seed);
}

void perform_gauge_smear_quda_(int *type, int *n_steps, double *coeff,
                               int *ierr) {
  *ierr = performGaugeSmearQuda((QudaAmdGaugeSmearType)*type, *n_steps,
                                *coeff);
}

void gauge_observables_quda_(double *plaq, double *qcharge, double *energy,
                             int *ierr) {
  *ierr = gaugeObservablesQuda(plaq, qcharge, energy);
}

void project_su3_quda_(int *ierr) { *ierr = projectSU3Quda(); }

void compute_gauge_fixing_ovr_quda_(int *gauge_dir, int *max_iter,
                                    double *tol, int *ierr) {
  *ierr = computeGaugeFixingOVRQuda(*gauge_dir, *max_iter, *tol);
}

void compute_gauge_fixing_fft_quda_(int *gauge_dir, int *max_iter,
                                    double *alpha, double *tol, int *ierr) {
  *ierr = computeGaugeFixingFFTQuda(*gauge_dir, *max_iter, *alpha, *tol);
}

void contract_quda_(void *h_out, void *h_x, void *h_y,
                    QudaAmdInvertParam *param, int *mode, int *ierr) {
  *ierr = contractQuda(h_out, h_x, h_y, param, *mode);
}

void invert_multi_shift_quda_(void **h_x, void *h_b,
                              QudaAmdInvertParam *param, double *offsets,
                              int *num_offset, int *ierr) {
  *ierr = invertMultiShiftQuda(h_x, h_b, param, offsets, *num_offset);
}

}  // extern "C"
