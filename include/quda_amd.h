/* quda_amd C API — the extern "C" ABI of the MI355X-native lattice-QCD
 * framework (role of the reference's include/quda.h:933-1808 entry points
 * and its QudaGaugeParam/QudaInvertParam structs, re-derived for this
 * engine; field names follow the public quda.h conventions so client code
 * ports by renaming the struct types).
 *
 * Data layouts (host side, always complex double unless stated):
 *   gauge : contiguous [4 dirs][2 parities][Vcb][3][3], even sites first,
 *           x fastest within a parity (the QDP-style even-odd order).
 *   spinor: contiguous [2 parities][Vcb][4 spins][3 colors]
 *           (staggered: [2][Vcb][3]).
 *
 * Link against libquda_amd_c.so. The library embeds the engine; call
 * initQuda() once before anything else. Set QUDA_AMD_DEVICE=cpu to force
 * host execution (tests), default is the current HIP device.
 */
#ifndef QUDA_AMD_H
#define QUDA_AMD_H

#ifdef __cplusplus
extern "C" {
#endif

typedef enum { QUDA_AMD_DOUBLE = 8, QUDA_AMD_SINGLE = 4, QUDA_AMD_HALF = 2 } QudaAmdPrecision;
typedef enum { QUDA_AMD_RECON_NO = 18, QUDA_AMD_RECON_12 = 12, QUDA_AMD_RECON_8 = 8 } QudaAmdReconstruct;
typedef enum { QUDA_AMD_PERIODIC_T = 1, QUDA_AMD_ANTI_PERIODIC_T = -1 } QudaAmdTboundary;
typedef enum {
  QUDA_AMD_WILSON_DSLASH = 0,
  QUDA_AMD_CLOVER_WILSON_DSLASH = 1,
  QUDA_AMD_TWISTED_MASS_DSLASH = 2,
  QUDA_AMD_TWISTED_CLOVER_DSLASH = 3,
  QUDA_AMD_STAGGERED_DSLASH = 4,
  QUDA_AMD_ASQTAD_DSLASH = 5,
  QUDA_AMD_DOMAIN_WALL_DSLASH = 6,
  QUDA_AMD_MOBIUS_DWF_DSLASH = 7,
  QUDA_AMD_LAPLACE_DSLASH = 8
} QudaAmdDslashType;
typedef enum {
  QUDA_AMD_CG_INVERTER = 0,
  QUDA_AMD_BICGSTAB_INVERTER = 1,
  QUDA_AMD_GCR_INVERTER = 2,
  QUDA_AMD_MR_INVERTER = 3,
  QUDA_AMD_BICGSTABL_INVERTER = 4,
  QUDA_AMD_CA_CG_INVERTER = 5,
  QUDA_AMD_CA_GCR_INVERTER = 6,
  QUDA_AMD_CGNE_INVERTER = 7,
  QUDA_AMD_CGNR_INVERTER = 8
} QudaAmdInverterType;
typedef enum {
  QUDA_AMD_MAT_SOLUTION = 0,     /* solve M x = b on the full lattice   */
  QUDA_AMD_MATPC_SOLUTION = 1    /* even-odd preconditioned solve       */
} QudaAmdSolutionType;

typedef struct QudaAmdGaugeParam_s {
  int X[4];                      /* local lattice extents               */
  QudaAmdPrecision cpu_prec;     /* host data precision (double only)   */
  QudaAmdPrecision cuda_prec;
  QudaAmdPrecision cuda_prec_sloppy;
  QudaAmdReconstruct reconstruct;
  QudaAmdReconstruct reconstruct_sloppy;
  double anisotropy;
  QudaAmdTboundary t_boundary;
  int staggered_phase_applied;   /* input links already carry eta(x)    */
} QudaAmdGaugeParam;

typedef struct QudaAmdInvertParam_s {
  QudaAmdDslashType dslash_type;
  QudaAmdInverterType inv_type;
  QudaAmdSolutionType solution_type;
  double kappa;
  double mass;                   /* staggered */
  double mu;                     /* twisted   */
  double clover_csw;             /* used by loadCloverQuda(NULL, ...)   */
  double tol;
  int maxiter;
  double reliable_delta;
  QudaAmdPrecision cpu_prec;     /* host data precision (double only)   */
  QudaAmdPrecision cuda_prec;
  QudaAmdPrecision cuda_prec_sloppy;
  int Ls;                        /* domain wall */
  double m5, b5, c5;
  /* output fields, filled by invertQuda */
  int iter;
  double true_res;
  double secs;
  double gflops;
} QudaAmdInvertParam;

/* default-initialized params (role of newQudaGaugeParam/newQudaInvertParam) */
QudaAmdGaugeParam newQudaAmdGaugeParam(void);
QudaAmdInvertParam newQudaAmdInvertParam(void);

/* lifecycle (ref: initQuda interface_quda.cpp:522 / endQuda) */
int initQuda(int device);
int endQuda(void);

/* resident gauge/clover management (ref: loadGaugeQuda
 * interface_quda.cpp:571, loadCloverQuda, freeGaugeQuda) */
int loadGaugeQuda(const void *h_gauge, QudaAmdGaugeParam *param);
int saveGaugeQuda(void *h_gauge, QudaAmdGaugeParam *param);
int freeGaugeQuda(void);
/* h_clover == NULL: compute the clover term from the resident gauge with
 * param->clover_csw (QUDA's compute_clover path) */
int loadCloverQuda(const void *h_clover, const void *h_clovinv,
                   QudaAmdInvertParam *param);

/* solves & operator applications (ref: invertQuda interface_quda.cpp:2986,
 * dslashQuda:1709, MatQuda, MatDagMatQuda, invertMultiShiftQuda:3405) */
int invertQuda(void *h_x, const void *h_b, QudaAmdInvertParam *param);
int invertMultiShiftQuda(void **h_x, const void *h_b,
                         QudaAmdInvertParam *param, const double *offsets,
                         int num_offset);
int dslashQuda(void *h_out, const void *h_in, QudaAmdInvertParam *param,
               int parity);
int MatQuda(void *h_out, const void *h_in, QudaAmdInvertParam *param);
int MatDagMatQuda(void *h_out, const void *h_in, QudaAmdInvertParam *param);

/* observables (ref: plaqQuda quda.h:1462; plaq[0]=total, [1]=spatial,
 * [2]=temporal) */
int plaqQuda(double plaq[3]);

/* last error message ("" when the previous call succeeded) */
const char *qudaAmdLastError(void);

#ifdef __cplusplus
}
#endif
#endif /* QUDA_AMD_H */
