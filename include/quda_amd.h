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
  /* opaque MG handle from newMultigridQuda (NULL = unpreconditioned);
   * consumed by invertQuda with inv_type GCR (ref: the
   * QudaInvertParam::preconditioner field of quda.h) */
  void *preconditioner;
} QudaAmdInvertParam;

/* eigensolver request (ref: QudaEigParam quda.h:471, consumed subset) */
typedef struct QudaAmdEigParam_s {
  int n_ev;                      /* eigenpairs wanted                   */
  int n_kr;                      /* Krylov/Lanczos subspace size        */
  double tol;
  int max_restarts;
  int use_norm_op;               /* 1: TRLM on MdagM; 0: IRAM on M      */
  int use_poly_acc;              /* Chebyshev acceleration (TRLM)       */
  int poly_deg;
  double a_min, a_max;           /* Chebyshev window                    */
  int spectrum_largest;          /* 0 = smallest end, 1 = largest       */
} QudaAmdEigParam;

/* multigrid setup request (ref: QudaMultigridParam quda.h:570 subset) */
typedef struct QudaAmdMultigridParam_s {
  int geo_block_size[4];         /* aggregation block                   */
  int n_vec;                     /* near-null vectors                   */
  int n_level;                   /* 2 or 3                              */
} QudaAmdMultigridParam;

/* smearing kinds for performGaugeSmearQuda (ref: QudaGaugeSmearType) */
typedef enum {
  QUDA_AMD_SMEAR_APE = 0,
  QUDA_AMD_SMEAR_STOUT = 1,
  QUDA_AMD_SMEAR_WILSON_FLOW = 2,
  QUDA_AMD_SMEAR_HYP = 3
} QudaAmdGaugeSmearType;

/* default-initialized params (role of newQudaGaugeParam/newQudaInvertParam) */
QudaAmdGaugeParam newQudaAmdGaugeParam(void);
QudaAmdInvertParam newQudaAmdInvertParam(void);
QudaAmdEigParam newQudaAmdEigParam(void);
QudaAmdMultigridParam newQudaAmdMultigridParam(void);

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

/* eigensolve the resident operator described by ip (ref: eigensolveQuda
 * interface_quda.cpp:2524): fills evals_re/evals_im[n_ev]; h_evecs may be
 * NULL or an array of ep->n_ev host spinor buffers */
int eigensolveQuda(double *evals_re, double *evals_im, void **h_evecs,
                   QudaAmdInvertParam *ip, QudaAmdEigParam *ep);

/* multigrid lifecycle (ref: newMultigridQuda interface_quda.cpp:2772,
 * updateMultigridQuda, destroyMultigridQuda). The returned handle plugs
 * into QudaAmdInvertParam.preconditioner for GCR solves. */
void *newMultigridQuda(QudaAmdInvertParam *ip, QudaAmdMultigridParam *mp);
int updateMultigridQuda(void *mg, QudaAmdInvertParam *ip);
int destroyMultigridQuda(void *mg);

/* HMC surface (ref: computeGaugeForceQuda, updateGaugeFieldQuda,
 * momActionQuda, momResidentQuda, gaussMomQuda). Momentum layout matches
 * the gauge layout: [4][2][Vcb][3][3] complex double (antihermitian). */
int computeGaugeForceQuda(void *h_mom, double beta);
int updateGaugeFieldQuda(const void *h_mom, double dt);
int momActionQuda(double *action, const void *h_mom);
int momResidentQuda(const void *h_mom);      /* NULL clears              */
int gaussMomQuda(void *h_mom, long seed);

/* gauge smearing of the resident field in place (ref:
 * performGaugeSmearQuda / performWFlowQuda; coeff = alpha/rho/epsilon) */
int performGaugeSmearQuda(QudaAmdGaugeSmearType type, int n_steps,
                          double coeff);

/* combined gauge observables on the resident field (ref:
 * gaugeObservablesQuda; energy[0]=plaq-based E, energy[1]=clover E) */
int gaugeObservablesQuda(double plaq[3], double *qcharge, double energy[2]);

/* project the resident links back onto SU(3) (ref: projectSU3Quda) */
int projectSU3Quda(void);

/* gauge fixing of the resident field (ref: computeGaugeFixingOVRQuda /
 * computeGaugeFixingFFTQuda; gauge_dir 4 = Landau, 3 = Coulomb) */
int computeGaugeFixingOVRQuda(int gauge_dir, int max_iter, double tol);
int computeGaugeFixingFFTQuda(int gauge_dir, int max_iter, double alpha,
                              double tol);

/* open-spin (mode 0) / DeGrand-Rossi gamma-insertion (mode 1) contraction
 * of two host propagator fields; out: [V][4][4] (mode 0) or [V][16]
 * (mode 1) complex double in lexicographic site order (ref: contractQuda) */
int contractQuda(void *h_out, const void *h_x, const void *h_y,
                 QudaAmdInvertParam *param, int mode);

/* last error message ("" when the previous call succeeded) */
const char *qudaAmdLastError(void);

#ifdef __cplusplus
}
#endif
#endif /* QUDA_AMD_H */
