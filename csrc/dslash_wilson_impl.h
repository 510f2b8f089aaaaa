#include <cstdlib>
// Shared runtime->template dispatch for the Wilson dslash TUs.
#pragma once
#include "dslash_wilson.h"
#include "launchers.h"

template <typename Prec, int RECON>
static void dslash_launch_all(const DslashCall &c, hipStream_t st) {
  using R = typename Prec::Real;
  // BlasField.Vcb carries the CHUNK STRIDE (= Ls*Vcb for a 5-d field whose
  // s-slice is addressed via a pointer offset); c.Vcb is the 4-d cb volume
  SpinorAcc<Prec> out{(typename Prec::Store *)c.out.data, (float *)c.out.norm, c.out.Vcb};
  SpinorAcc<Prec> in{(typename Prec::Store *)c.in.data, (float *)c.in.norm, c.in.Vcb};
  SpinorAcc<Prec> x{(typename Prec::Store *)c.x.data, (float *)c.x.norm, c.x.Vcb};
  // parity-offset the stencil gauge base: [2][NCH][V][W]
  const long gpar = (long)GaugeAcc<Prec, RECON>::NCH * c.Vcb * Prec::W;
  const auto *g0 = (const typename Prec::Store *)c.gauge;
  GaugeAcc<Prec, RECON> g{g0 + c.parity * gpar, g0 + (1 - c.parity) * gpar,
                          c.Vcb};
  CloverAcc<Prec> cl{(const typename Prec::Store *)c.clover, c.Vcb};
  LatDims d{{c.Xdim[0], c.Xdim[1], c.Xdim[2], c.Xdim[3]}, c.parity_offset, c.Vcb};
  GhostAcc<Prec> gh{};
  gh.mask = c.comm_mask;
  for (int k = 0; k < 8; ++k) {
    gh.buf[k] = (const typename Prec::Store *)c.ghost[k];
    gh.nrm[k] = c.ghost_nrm[k];
  }
  for (int k = 0; k < 4; ++k) gh.Fcb[k] = c.face_cb[k];
  int blk = qa_dslash_block();
  int grid = (int)((c.Vcb + blk - 1) / blk);
  R a = (R)c.a;
  R br = (R)c.b_re, bi = (R)c.b_im;
  long n_ext = 0;
  for (int m = 0; m < 4; ++m)
    if ((c.comm_mask >> m) & 1) n_ext += 2 * c.face_cb[m];
  int grid_ext = (int)((n_ext + blk - 1) / blk);

  int waves = qa_dslash_waves();
  int grid64 = (int)((c.Vcb + 63) / 64);

  // LDS-tiled path (half/quarter, local, 4-d, tile-divisible dims,
  // PLAIN/CLOV_POST/TWIST_POST): see k_dslash_wilson_lds
  if constexpr (Prec::has_norm) {
    bool lds_ok = qa_dslash_lds() && c.kt == 0 && c.comm_mask == 0 &&
                  c.in.Vcb == c.Vcb && c.out.Vcb == c.Vcb &&
                  (c.mode == PLAIN || c.mode == CLOV_POST ||
                   c.mode == TWIST_POST) &&
                  c.Xdim[0] % LdsTile::BX == 0 &&
                  c.Xdim[1] % LdsTile::BY == 0 &&
                  c.Xdim[2] % LdsTile::BZ == 0 && c.Xdim[3] % LdsTile::BT == 0;
    if (lds_ok) {
      long ntile = ((long)c.Xdim[0] / LdsTile::BX) * (c.Xdim[1] / LdsTile::BY) *
                   (c.Xdim[2] / LdsTile::BZ) * (c.Xdim[3] / LdsTile::BT);
#define QA_LDS_LAUNCH(DAG, MODE, XPAY)                                              hipLaunchKernelGGL((k_dslash_wilson_lds<Prec, RECON, DAG, MODE, XPAY>),                          dim3((unsigned)ntile), dim3(256), 0, st, out, in, g,                          cl, d, c.parity, a, x, br, bi)
#define QA_LDS_MODES(DAG)                                                           switch (c.mode) {                                                               case PLAIN:                                                                     if (c.xpay) QA_LDS_LAUNCH(DAG, PLAIN, true);                                  else QA_LDS_LAUNCH(DAG, PLAIN, false);                                        break;                                                                      case CLOV_POST:                                                                 if (c.xpay) QA_LDS_LAUNCH(DAG, CLOV_POST, true);                              else QA_LDS_LAUNCH(DAG, CLOV_POST, false);                                    break;                                                                      case TWIST_POST:                                                                if (c.xpay) QA_LDS_LAUNCH(DAG, TWIST_POST, true);                             else QA_LDS_LAUNCH(DAG, TWIST_POST, false);                                   break;                                                                    }
      if (!c.dagger) { QA_LDS_MODES(false) } else { QA_LDS_MODES(true) }
#undef QA_LDS_MODES
#undef QA_LDS_LAUNCH
      return;
    }
  }

#define QA_LAUNCH(DAG, MODE, XPAY, KT)                                        \
  if (c.kt == 3)                                                              \
    hipLaunchKernelGGL((k_dslash_wilson_exterior<Prec, RECON, DAG, MODE, XPAY>), \
                       dim3(grid_ext), dim3(blk), 0, st, out, in, g, cl, d,   \
                       c.parity, a, x, gh, n_ext, br, bi);                    \
  else if (waves == 3)                                                        \
    hipLaunchKernelGGL((k_dslash_wilson<Prec, RECON, DAG, MODE, XPAY, KT, 3>),\
                       dim3(grid64), dim3(64), 0, st, out, in, g, cl, d,      \
                       c.parity, a, x, gh, br, bi);                           \
  else                                                                        \
    hipLaunchKernelGGL((k_dslash_wilson<Prec, RECON, DAG, MODE, XPAY, KT>),   \
                       dim3(grid), dim3(blk), 0, st, out, in, g, cl, d,       \
                       c.parity, a, x, gh, br, bi)

#define QA_MODES(DAG, KT)                                                      \
  switch (c.mode) {                                                            \
    case PLAIN:                                                                \
      if (c.xpay) QA_LAUNCH(DAG, PLAIN, true, KT);                             \
      else QA_LAUNCH(DAG, PLAIN, false, KT);                                   \
      break;                                                                   \
    case CLOV_POST:                                                            \
      if (c.xpay) QA_LAUNCH(DAG, CLOV_POST, true, KT);                         \
      else QA_LAUNCH(DAG, CLOV_POST, false, KT);                               \
      break;                                                                   \
    case CLOV_X: QA_LAUNCH(DAG, CLOV_X, true, KT); break;                      \
    case TWIST_POST:                                                           \
      if (c.xpay) QA_LAUNCH(DAG, TWIST_POST, true, KT);                        \
      else QA_LAUNCH(DAG, TWIST_POST, false, KT);                              \
      break;                                                                   \
    case TWIST_X: QA_LAUNCH(DAG, TWIST_X, true, KT); break;                    \
    case CLOVTW_X: QA_LAUNCH(DAG, CLOVTW_X, true, KT); break;                  \
  }

#define QA_DISPATCH(KT)                                                        \
  if (!c.dagger) { QA_MODES(false, KT) } else { QA_MODES(true, KT) }

  switch (c.kt) {
    case 0: { QA_DISPATCH(KT_LOCAL) } break;
    case 1: { QA_DISPATCH(KT_FUSED) } break;
    case 2: { QA_DISPATCH(KT_INTERIOR) } break;
    case 3: { QA_DISPATCH(KT_LOCAL) } break;  // KT arg unused for exterior
  }
#undef QA_DISPATCH
#undef QA_MODES
#undef QA_LAUNCH
}

template <typename Prec>
static void pack_launch(const PackCall &c, hipStream_t st) {
  SpinorAcc<Prec> in{(typename Prec::Store *)c.in.data, (float *)c.in.norm, c.in.Vcb};
  LatDims d{{c.Xdim[0], c.Xdim[1], c.Xdim[2], c.Xdim[3]}, c.parity_offset, c.Vcb};
  int blk = 256;
  int grid = (int)((c.Fcb + blk - 1) / blk);
  auto *dst = (typename Prec::Store *)c.dst;

#define QA_PACK(MU, S01, EDGE)                                                 \
  hipLaunchKernelGGL((k_pack_face<Prec, MU, S01, EDGE>), dim3(grid), dim3(blk),\
                     0, st, dst, c.dst_nrm, in, d, c.parity, c.Fcb)
#define QA_PACK_MU(MU)                                                         \
  case MU:                                                                     \
    if (c.s01 == 0) { if (c.edge) QA_PACK(MU, 0, true); else QA_PACK(MU, 0, false); } \
    else            { if (c.edge) QA_PACK(MU, 1, true); else QA_PACK(MU, 1, false); } \
    break;

  switch (c.mu) {
    QA_PACK_MU(0)
    QA_PACK_MU(1)
    QA_PACK_MU(2)
    QA_PACK_MU(3)
  }
#undef QA_PACK_MU
#undef QA_PACK
}
