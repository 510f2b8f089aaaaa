// C++ launcher API between the HIP kernel TUs and bindings.cpp.
#pragma once
#include <hip/hip_runtime.h>

#include <cstdlib>

// runtime-tunable workgroup size (set by the Python autotuner via the
// set_dslash_block binding; env QUDA_AMD_DSLASH_BLOCK seeds the default)
inline int &qa_dslash_block_ref() {
  static int blk = [] {
    const char *e = getenv("QUDA_AMD_DSLASH_BLOCK");
    int v = e ? atoi(e) : 64;  // 1 wave/WG measures best (see profiles/)
    return (v == 64 || v == 128 || v == 256) ? v : 64;
  }();
  return blk;
}
inline int qa_dslash_block() { return qa_dslash_block_ref(); }

// occupancy experiment: 0 = default (256-thread cap, ~2 waves/SIMD at the
// measured 222 VGPRs), 3 = 64-thread workgroups with __launch_bounds__
// forcing >= 3 waves/SIMD (VGPR cap 168). Only {0,3} are compiled.
inline int &qa_dslash_waves_ref() {
  static int w = [] {
    const char *e = getenv("QUDA_AMD_DSLASH_WAVES");
    int v = e ? atoi(e) : 0;
    return v == 3 ? 3 : 0;
  }();
  return w;
}
inline int qa_dslash_waves() { return qa_dslash_waves_ref(); }

// LDS-tiled dslash policy (0 = off, 1 = on where eligible: half/quarter,
// local, 4-d, tile-divisible dims; set_dslash_lds binding / autotuner)
inline int &qa_dslash_lds_ref() {
  static int v = []() {
    const char *e = std::getenv("QUDA_AMD_DSLASH_LDS");
    return e ? std::atoi(e) : 0;
  }();
  return v;
}
inline int qa_dslash_lds() { return qa_dslash_lds_ref(); }

struct BlasField {
  void *data;
  void *norm;  // nullptr unless half
  long Vcb;    // chunk stride (cb volume)
};

enum BlasOp {
  BLAS_AXPY = 0,
  BLAS_AXPY_NORM2,
  BLAS_XPAY,
  BLAS_AXPBY,
  BLAS_CAXPY,
  BLAS_XMY_NORM2,
  BLAS_SCAL,
  BLAS_NORM2,
  BLAS_REDOT,
  BLAS_CDOT,
  BLAS_CAXPBY,
  BLAS_TRIPLE_CG,  // x += a p; r -= a Ap; -> ||r||^2 (ref tripleCGUpdate)
};

struct BlasCall {
  int op;        // BlasOp
  int prec;      // 0 double, 1 single, 2 half
  double a, b;   // scalars (caxpy: a=re, b=im)
  double c, d;   // second complex scalar (caxpby: b=(c,d))
  BlasField x, y;
  BlasField z, w;  // extra operands (TRIPLE_CG: x=p, y=Ap, z=x, w=r)
  long sites;    // npar * Vcb
  int ncomp;     // reals per site: 24 (Wilson) or 6 (staggered)
  bool det;      // deterministic reduce: result = per-block partials
  double *result;  // device ptr (2 doubles, or [grid][2] partials if det)
};

void launch_blas(const BlasCall &c, hipStream_t st);
void launch_convert(const BlasField &dst, int pdst, const BlasField &src,
                    int psrc, long sites, int ncomp, hipStream_t st);

// ---------------------------------------------------------------------------
struct DslashCall {
  // single-parity field views (data points at the parity slice)
  BlasField out, in, x;    // x used when xpay / CLOV_X
  const void *gauge;       // [mu][par][nch][Vcb][2]
  const void *clover;      // packed [par][nch][Vcb][W] (A or A^-1 slice)
  int Xdim[4];
  int parity_offset;
  long Vcb;
  int parity;
  bool dagger;
  int mode;   // 0 PLAIN, 1 CLOV_POST, 2 CLOV_X
  bool xpay;
  double a;
  double b_re, b_im;  // twist scalar for TWIST_*/CLOVTW_* modes
  int recon;  // 18 or 12
  // halo: ghost recv buffers per [2*mu+dir] (dir 1 = from +mu neighbor);
  // null when mask bit mu unset. comm_mask==0 => pure-local kernel.
  const void *ghost[8];
  const float *ghost_nrm[8];
  long face_cb[4];
  int comm_mask;
  int kt;  // 0 local, 1 fused, 2 interior, 3 exterior (csrc/dslash_wilson.h)
};

void launch_dslash_wilson_double(const DslashCall &c, hipStream_t st);
void launch_dslash_wilson_single(const DslashCall &c, hipStream_t st);
void launch_dslash_wilson_half(const DslashCall &c, hipStream_t st);
void launch_dslash_wilson_quarter(const DslashCall &c, hipStream_t st);

// ---------------------------------------------------------------------------
// multi-RHS Wilson(-clover): NRHS in {2,4} sides share each gauge/clover
// load (csrc/dslash_wilson_mrhs.h). mode PLAIN or CLOV_POST only.
#define QA_MRHS_MAX 4
struct DslashMrhsCall {
  int nrhs;
  BlasField out[QA_MRHS_MAX], in[QA_MRHS_MAX], x[QA_MRHS_MAX];
  const void *gauge;
  const void *clover;
  int Xdim[4];
  int parity_offset;
  long Vcb;
  int parity;
  bool dagger;
  int mode;
  bool xpay;
  double a;
  int recon;
  // batched ghosts: slice-0 base per [2*mu+dir]; kernel offsets by rhs
  const void *ghost[8];
  const float *ghost_nrm[8];
  long face_cb[4];
  int comm_mask;
  int kt;  // 0 local, 1 fused, 2 interior (exterior runs per-RHS)
  int prec;
};
void launch_dslash_wilson_mrhs(const DslashMrhsCall &c, hipStream_t st);

// pack one (mu, s01, edge) face of `in` into dst (see csrc/halo.h)
struct PackCall {
  BlasField in;       // dslash input spinor (single parity view)
  void *dst;
  float *dst_nrm;     // half only
  int Xdim[4];
  int parity_offset;
  long Vcb;
  int parity;         // parity of `in`
  int mu;
  int s01;            // projector sign index (dir XOR dagger)
  int edge;           // 0: x_mu = 0 face, 1: x_mu = X-1 face
  long Fcb;
  int prec;
  int depth = 1;      // ghost layers (staggered Naik = 3)
};
void launch_pack_face_double(const PackCall &c, hipStream_t st);
void launch_pack_face_single(const PackCall &c, hipStream_t st);
void launch_pack_face_half(const PackCall &c, hipStream_t st);
void launch_pack_face_quarter(const PackCall &c, hipStream_t st);

struct CloverApplyCall {
  BlasField out, in;
  const void *clover;  // full [2][nch][Vcb][W]
  int parity;          // which parity slot of the clover field
  long Vcb;
  long sites;          // = Vcb (single parity)
  int prec;
};
void launch_clover_apply(const CloverApplyCall &c, hipStream_t st);

struct TwistApplyCall {
  BlasField out, in;
  double b_re, b_im;
  long sites;
  long tau3_vcb;  // >0: flavor-doublet g5*tau3 mode (flavor = site/tau3_vcb)
  bool acc;       // accumulate into out instead of overwrite
  int prec;
};
void launch_twist_apply(const TwistApplyCall &c, hipStream_t st);

// ---------------------------------------------------------------------------
struct StagDslashCall {
  BlasField out, in, x;  // single-parity views
  const void *gauge;     // stencil layout, like DslashCall
  const void *long_gauge;  // Naik links (stencil layout, shift=3, recon 18); null = naive
  int Xdim[4];
  int parity_offset;
  long Vcb;
  int parity;
  bool xpay;
  double a, b;  // out = [a*x +] b*(D in); dagger folds into b
  int recon;
  const void *ghost[8];
  const float *ghost_nrm[8];
  long face_cb[4];
  int comm_mask;
  int kt;  // 0 local, 1 fused, 2 interior, 3 exterior
  int prec;
  int ghost_depth;  // 1, or 3 when long links cross rank boundaries
};
void launch_dslash_staggered(const StagDslashCall &c, hipStream_t st);
void launch_pack_face_stag(const PackCall &c, hipStream_t st);

// ---------------------------------------------------------------------------
struct Dwf5Call {
  BlasField out, in, x;  // 5-d single-parity fields (Vcb = chunk stride Ls*Vcb4)
  long Vcb4;
  int Ls;
  bool xpay;
  bool dagger;
  double a, alpha, beta, mf;
  int prec;
  int kind;  // 0 = dslash5 (Ds apply), 1 = m5inv
};
void launch_dwf5(const Dwf5Call &c, hipStream_t st);

// ---------------------------------------------------------------------------
// coarse-grid dslash on MFMA (csrc/coarse.hip)
struct CoarseMfmaCall {
  const void *mats;  // [9][Na][Nc][Nc] complex64
  const void *nbr9;  // [Na][9] int64 source sites (>=Na -> ghost rows)
  const void *c;     // [Na + n_ghost][Nc][NR] complex64
  void *out;         // [Na][Nc][NR] complex64
  long Na;
  int Nc;            // multiple of 16
  int NR;            // <= 16
};
void launch_coarse_dslash_mfma(const CoarseMfmaCall &c, hipStream_t st);

// ---------------------------------------------------------------------------
// native heatbath / overrelaxation sweep (csrc/heatbath.hip)
struct HeatbathCall {
  void *u;  // [4][2][Vcb][3][3] complex double (oracle layout)
  int Xdim[4];
  int parity_offset;
  long Vcb;
  int parity;
  int mu;
  double beta_eff;
  unsigned long long seed;
  int mode;  // 0 heatbath, 1 overrelax
};
void launch_heatbath(const HeatbathCall &c, hipStream_t st);

struct StoutCall {
  void *out;       // [4][2][Vcb][3][3] complex double
  const void *in;
  const void *aux;  // expmul: the combined Z tensor
  int Xdim[4];
  int parity_offset;
  long Vcb;
  int mu;
  double rho;      // zmat: the eps scale
};
void launch_stout(const StoutCall &c, hipStream_t st);
void launch_zmat(const StoutCall &c, hipStream_t st);    // Z = eps TA[S U^d]
void launch_expmul(const StoutCall &c, hipStream_t st);  // U' = exp(Zc) U

#define QA_ZMAX 32

struct ZCoef {  // host-filled complex coefficient tables (double re/im)
  // zdslash5: out(s)_upper = au[s]*in(s) + wu[s]*in(su[s]); lower likewise
  double au[QA_ZMAX][2], al[QA_ZMAX][2];
  double wu[QA_ZMAX][2], wl[QA_ZMAX][2];
  int su[QA_ZMAX], sl[QA_ZMAX];
  // zm5inv: per chirality a sequence order ord[i] = physical slice of step
  // i and the recursion y[i] = di[i] * (r[i] - e[i] y[i-1]) with di = 1/d;
  // Sherman-Morrison corner weight cw couples step 0 to the LAST step.
  int ord_u[QA_ZMAX], ord_l[QA_ZMAX];
  double diu[QA_ZMAX][2], eu[QA_ZMAX][2], cwu[2];
  double dil[QA_ZMAX][2], el[QA_ZMAX][2], cwl[2];
};

struct ZDwf5Call {
  BlasField out, in, x;
  long Vcb4;
  int Ls;
  bool xpay;
  double a_re, a_im;
  int prec;
  int kind;  // 0 = zdslash5, 1 = zm5inv (double/single only)
  const ZCoef *zc;  // HOST pointer; launcher stages to device
};
void launch_zdwf5(const ZDwf5Call &c, hipStream_t st);

struct EofaVec;  // defined in dslash_dwf.h
struct Eofa5Call {
  BlasField out, in, x;
  long Vcb4;
  int Ls;
  bool xpay;
  bool dagger;
  double a, alpha, beta, mf;
  double u[32], w[32];  // see EofaVec semantics (m5inv: u = B^-1 u, sh/denom)
  double sh;
  int pm;
  int prec;
  int kind;  // 0 = m5_eofa, 1 = m5inv_eofa
};
void launch_eofa5(const Eofa5Call &c, hipStream_t st);
