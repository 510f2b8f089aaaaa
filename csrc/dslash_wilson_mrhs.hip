// Multi-RHS Wilson dslash TU (all precisions; NRHS 2 and 4).
#include "dslash_wilson_mrhs.h"
#include "launchers.h"

template <typename Prec, int RECON, int NRHS>
static void mrhs_launch(const DslashMrhsCall &c, hipStream_t st) {
  using S = typename Prec::Store;
  MrhsPtrs<Prec, NRHS> ptr;
  for (int r = 0; r < NRHS; ++r) {
    ptr.out[r] = (S *)c.out[r].data;
    ptr.out_n[r] = (float *)c.out[r].norm;
    ptr.in[r] = (const S *)c.in[r].data;
    ptr.in_n[r] = (const float *)c.in[r].norm;
    ptr.x[r] = (const S *)c.x[r].data;
    ptr.x_n[r] = (const float *)c.x[r].norm;
  }
  long Vs = c.out[0].Vcb;  // chunk stride
  const long gpar = (long)GaugeAcc<Prec, RECON>::NCH * c.Vcb * Prec::W;
  const auto *g0 = (const S *)c.gauge;
  GaugeAcc<Prec, RECON> g{g0 + c.parity * gpar, g0 + (1 - c.parity) * gpar,
                          c.Vcb};
  CloverAcc<Prec> cl{(const S *)c.clover, c.Vcb};
  LatDims d{{c.Xdim[0], c.Xdim[1], c.Xdim[2], c.Xdim[3]}, c.parity_offset,
            c.Vcb};
  GhostAcc<Prec> gh{};
  gh.mask = c.comm_mask;
  for (int k = 0; k < 8; ++k) {
    gh.buf[k] = (const S *)c.ghost[k];
    gh.nrm[k] = c.ghost_nrm[k];
  }
  for (int k = 0; k < 4; ++k) gh.Fcb[k] = c.face_cb[k];
  int blk = qa_dslash_block();
  int grid = (int)((c.Vcb + blk - 1) / blk);
  typename Prec::Real a = (typename Prec::Real)c.a;

#define QA_M_LAUNCH(DAG, MODE, XPAY, KT)                                      \
  hipLaunchKernelGGL(                                                         \
      (k_dslash_wilson_mrhs<Prec, RECON, DAG, MODE, XPAY, KT, NRHS>),         \
      dim3(grid), dim3(blk), 0, st, ptr, Vs, g, cl, d, c.parity, a, gh)

#define QA_M_MODES(DAG, KT)                                                   \
  if (c.mode == PLAIN) {                                                      \
    if (c.xpay) QA_M_LAUNCH(DAG, PLAIN, true, KT);                            \
    else QA_M_LAUNCH(DAG, PLAIN, false, KT);                                  \
  } else {                                                                    \
    if (c.xpay) QA_M_LAUNCH(DAG, CLOV_POST, true, KT);                        \
    else QA_M_LAUNCH(DAG, CLOV_POST, false, KT);                              \
  }

#define QA_M_KT(DAG)                                                          \
  switch (c.kt) {                                                             \
    case 0: QA_M_MODES(DAG, KT_LOCAL) break;                                  \
    case 1: QA_M_MODES(DAG, KT_FUSED) break;                                  \
    default: QA_M_MODES(DAG, KT_INTERIOR) break;                              \
  }

  if (!c.dagger) { QA_M_KT(false) } else { QA_M_KT(true) }
#undef QA_M_KT
#undef QA_M_MODES
#undef QA_M_LAUNCH
}

template <typename Prec>
static void mrhs_prec(const DslashMrhsCall &c, hipStream_t st) {
  if (c.recon == 12) {
    if (c.nrhs == 4) mrhs_launch<Prec, 12, 4>(c, st);
    else mrhs_launch<Prec, 12, 2>(c, st);
  } else {
    if (c.nrhs == 4) mrhs_launch<Prec, 18, 4>(c, st);
    else mrhs_launch<Prec, 18, 2>(c, st);
  }
}

void launch_dslash_wilson_mrhs(const DslashMrhsCall &c, hipStream_t st) {
  if (c.prec == 0) mrhs_prec<PrecDouble>(c, st);
  else if (c.prec == 1) mrhs_prec<PrecSingle>(c, st);
  else mrhs_prec<PrecHalf>(c, st);
}
