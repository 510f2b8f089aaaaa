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

}  // extern "C"
