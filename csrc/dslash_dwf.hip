// DWF/Moebius 5th-dimension TU (all precisions)
#include "dslash_dwf.h"
#include "launchers.h"

template <typename Prec>
static void dwf5_t(const Dwf5Call &c, hipStream_t st) {
  using R = typename Prec::Real;
  SpinorAcc<Prec> out{(typename Prec::Store *)c.out.data, (float *)c.out.norm, c.out.Vcb};
  SpinorAcc<Prec> in{(typename Prec::Store *)c.in.data, (float *)c.in.norm, c.in.Vcb};
  SpinorAcc<Prec> x{(typename Prec::Store *)c.x.data, (float *)c.x.norm, c.x.Vcb};
  int blk = 256;
  R a = (R)c.a, al = (R)c.alpha, be = (R)c.beta, mf = (R)c.mf;

#define QA_D5(XPAY, DAG)                                                      \
  if (c.kind == 0) {                                                          \
    long n = c.Vcb4 * c.Ls;                                                   \
    hipLaunchKernelGGL((k_dslash5<Prec, XPAY, DAG>),                          \
                       dim3((int)((n + blk - 1) / blk)), dim3(blk), 0, st,    \
                       out, in, x, c.Vcb4, c.Ls, a, al, be, mf);              \
  } else {                                                                    \
    hipLaunchKernelGGL((k_m5inv<Prec, XPAY, DAG>),                            \
                       dim3((int)((c.Vcb4 + blk - 1) / blk)), dim3(blk), 0,   \
                       st, out, in, x, c.Vcb4, c.Ls, a, al, be, mf);          \
  }

  if (c.xpay) { if (c.dagger) QA_D5(true, true) else QA_D5(true, false) }
  else        { if (c.dagger) QA_D5(false, true) else QA_D5(false, false) }
#undef QA_D5
}

void launch_dwf5(const Dwf5Call &c, hipStream_t st) {
  switch (c.prec) {
    case 0: dwf5_t<PrecDouble>(c, st); break;
    case 1: dwf5_t<PrecSingle>(c, st); break;
    case 2: dwf5_t<PrecHalf>(c, st); break;
    case 3: dwf5_t<PrecQuarter>(c, st); break;
  }
}

template <typename Prec>
static void zdwf5_t(const ZDwf5Call &c, const ZCoef *dzc, hipStream_t st) {
  using R = typename Prec::Real;
  SpinorAcc<Prec> out{(typename Prec::Store *)c.out.data, (float *)c.out.norm, c.out.Vcb};
  SpinorAcc<Prec> in{(typename Prec::Store *)c.in.data, (float *)c.in.norm, c.in.Vcb};
  SpinorAcc<Prec> x{(typename Prec::Store *)c.x.data, (float *)c.x.norm, c.x.Vcb};
  int blk = 256;
  R ar = (R)c.a_re, ai = (R)c.a_im;
#define QA_Z5(XPAY)                                                           \
  if (c.kind == 0) {                                                          \
    long n = c.Vcb4 * c.Ls;                                                   \
    hipLaunchKernelGGL((k_zdslash5<Prec, XPAY>),                              \
                       dim3((int)((n + blk - 1) / blk)), dim3(blk), 0, st,    \
                       out, in, x, c.Vcb4, c.Ls, ar, ai, dzc);                \
  } else {                                                                    \
    hipLaunchKernelGGL((k_zm5inv<Prec, XPAY>),                                \
                       dim3((int)((c.Vcb4 + blk - 1) / blk)), dim3(blk), 0,   \
                       st, out, in, x, c.Vcb4, c.Ls, ar, ai, dzc);            \
  }
  if (c.xpay) QA_Z5(true) else QA_Z5(false)
#undef QA_Z5
}

void launch_zdwf5(const ZDwf5Call &c, hipStream_t st) {
  // stage the host-assembled table into a persistent device buffer
  // (stream-ordered: the async copy precedes the kernel on the same stream)
  static ZCoef *dzc = nullptr;
  if (!dzc) (void)hipMalloc(&dzc, sizeof(ZCoef));
  (void)hipMemcpyAsync(dzc, c.zc, sizeof(ZCoef), hipMemcpyHostToDevice, st);
  switch (c.prec) {
    case 0: zdwf5_t<PrecDouble>(c, dzc, st); break;
    case 1: zdwf5_t<PrecSingle>(c, dzc, st); break;
  }
}

template <typename Prec>
static void eofa5_t(const Eofa5Call &c, hipStream_t st) {
  using R = typename Prec::Real;
  SpinorAcc<Prec> out{(typename Prec::Store *)c.out.data, (float *)c.out.norm, c.out.Vcb};
  SpinorAcc<Prec> in{(typename Prec::Store *)c.in.data, (float *)c.in.norm, c.in.Vcb};
  SpinorAcc<Prec> x{(typename Prec::Store *)c.x.data, (float *)c.x.norm, c.x.Vcb};
  EofaVec ev{};
  for (int s_ = 0; s_ < c.Ls && s_ < QA_ZMAX; ++s_) {
    ev.u[s_] = c.u[s_];
    ev.w[s_] = c.w[s_];
  }
  ev.sh = c.sh;
  ev.pm = c.pm;
  int blk = 256;
  int grid = (int)((c.Vcb4 + blk - 1) / blk);
  R a = (R)c.a, al = (R)c.alpha, be = (R)c.beta, mf = (R)c.mf;
#define QA_E5(XPAY, DAG)                                                      \
  if (c.kind == 0)                                                            \
    hipLaunchKernelGGL((k_m5_eofa<Prec, XPAY, DAG>), dim3(grid), dim3(blk),   \
                       0, st, out, in, x, c.Vcb4, c.Ls, a, al, be, mf, ev);   \
  else                                                                        \
    hipLaunchKernelGGL((k_m5inv_eofa<Prec, XPAY, DAG>), dim3(grid),           \
                       dim3(blk), 0, st, out, in, x, c.Vcb4, c.Ls, a, al,     \
                       be, mf, ev);
  if (c.xpay) { if (c.dagger) QA_E5(true, true) else QA_E5(true, false) }
  else        { if (c.dagger) QA_E5(false, true) else QA_E5(false, false) }
#undef QA_E5
}

void launch_eofa5(const Eofa5Call &c, hipStream_t st) {
  switch (c.prec) {
    case 0: eofa5_t<PrecDouble>(c, st); break;
    case 1: eofa5_t<PrecSingle>(c, st); break;
    case 2: eofa5_t<PrecHalf>(c, st); break;
  }
}
