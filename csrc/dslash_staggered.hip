// staggered dslash TU (all precisions; kernel is small)
#include "dslash_staggered.h"
#include "launchers.h"

template <typename Prec, int RECON>
static void stag_launch(const StagDslashCall &c, hipStream_t st) {
  using R = typename Prec::Real;
  StagAcc<Prec> out{(typename Prec::Store *)c.out.data, (float *)c.out.norm, c.Vcb};
  StagAcc<Prec> in{(typename Prec::Store *)c.in.data, (float *)c.in.norm, c.Vcb};
  StagAcc<Prec> x{(typename Prec::Store *)c.x.data, (float *)c.x.norm, c.Vcb};
  const long gpar = (long)GaugeAcc<Prec, RECON>::NCH * c.Vcb * Prec::W;
  const auto *g0 = (const typename Prec::Store *)c.gauge;
  GaugeAcc<Prec, RECON> g{g0 + c.parity * gpar, g0 + (1 - c.parity) * gpar,
                          c.Vcb};
  const typename Prec::Store *l0 = (const typename Prec::Store *)c.long_gauge;
  const long lpar = (long)GaugeAcc<Prec, 18>::NCH * c.Vcb * Prec::W;
  // 3-hop neighbors flip parity too (odd shift): bwd long links also read
  // the other parity block's fwd slots
  GaugeAcc<Prec, 18> lng{l0 ? l0 + c.parity * lpar : nullptr,
                         l0 ? l0 + (1 - c.parity) * lpar : nullptr, c.Vcb};
  LatDims d{{c.Xdim[0], c.Xdim[1], c.Xdim[2], c.Xdim[3]}, c.parity_offset, c.Vcb};
  GhostAcc<Prec, 6> gh{};
  gh.mask = c.comm_mask;
  for (int k = 0; k < 8; ++k) {
    gh.buf[k] = (const typename Prec::Store *)c.ghost[k];
    gh.nrm[k] = c.ghost_nrm[k];
  }
  for (int k = 0; k < 4; ++k) gh.Fcb[k] = c.face_cb[k];
  gh.depth = c.ghost_depth > 0 ? c.ghost_depth : 1;
  int blk = 256;
  int grid = (int)((c.Vcb + blk - 1) / blk);
  R a = (R)c.a, b = (R)c.b;
  long n_ext = 0;
  for (int m = 0; m < 4; ++m)
    if ((c.comm_mask >> m) & 1) n_ext += 2 * c.face_cb[m];
  int grid_ext = (int)((n_ext + blk - 1) / blk);

#define QA_SLAUNCH(XPAY, KT)                                                   \
  if (c.long_gauge)                                                            \
    hipLaunchKernelGGL((k_dslash_staggered<Prec, RECON, XPAY, KT, true>),      \
                       dim3(grid), dim3(blk), 0, st, out, in, g, lng, d,       \
                       c.parity, a, b, x, gh);                                 \
  else                                                                         \
    hipLaunchKernelGGL((k_dslash_staggered<Prec, RECON, XPAY, KT, false>),     \
                       dim3(grid), dim3(blk), 0, st, out, in, g, lng, d,       \
                       c.parity, a, b, x, gh)
  if (c.kt == 3) {
    hipLaunchKernelGGL((k_dslash_staggered_exterior<Prec, RECON>),
                       dim3(grid_ext), dim3(blk), 0, st, out, in, g, d,
                       c.parity, b, gh, n_ext);
  } else if (c.kt == 0) {
    if (c.xpay) QA_SLAUNCH(true, KT_LOCAL); else QA_SLAUNCH(false, KT_LOCAL);
  } else if (c.kt == 1) {
    if (c.xpay) QA_SLAUNCH(true, KT_FUSED); else QA_SLAUNCH(false, KT_FUSED);
  } else {
    if (c.xpay) QA_SLAUNCH(true, KT_INTERIOR); else QA_SLAUNCH(false, KT_INTERIOR);
  }
#undef QA_SLAUNCH
}

template <typename Prec>
static void stag_recon(const StagDslashCall &c, hipStream_t st) {
  if (c.recon == 12) stag_launch<Prec, 12>(c, st);
  else stag_launch<Prec, 18>(c, st);
}

void launch_dslash_staggered(const StagDslashCall &c, hipStream_t st) {
  switch (c.prec) {
    case 0: stag_recon<PrecDouble>(c, st); break;
    case 1: stag_recon<PrecSingle>(c, st); break;
    case 2: stag_recon<PrecHalf>(c, st); break;
  }
}

template <typename Prec>
static void stag_pack(const PackCall &c, hipStream_t st) {
  StagAcc<Prec> in{(typename Prec::Store *)c.in.data, (float *)c.in.norm, c.Vcb};
  LatDims d{{c.Xdim[0], c.Xdim[1], c.Xdim[2], c.Xdim[3]}, c.parity_offset, c.Vcb};
  int blk = 256;
  int depth = c.depth > 0 ? c.depth : 1;
  long n = c.Fcb * depth;
  int grid = (int)((n + blk - 1) / blk);
  auto *dst = (typename Prec::Store *)c.dst;
  if (c.edge)
    hipLaunchKernelGGL((k_pack_face_stag<Prec, true>), dim3(grid), dim3(blk),
                       0, st, dst, c.dst_nrm, in, d, c.parity, c.mu, c.Fcb,
                       depth);
  else
    hipLaunchKernelGGL((k_pack_face_stag<Prec, false>), dim3(grid), dim3(blk),
                       0, st, dst, c.dst_nrm, in, d, c.parity, c.mu, c.Fcb,
                       depth);
}

void launch_pack_face_stag(const PackCall &c, hipStream_t st) {
  switch (c.prec) {
    case 0: stag_pack<PrecDouble>(c, st); break;
    case 1: stag_pack<PrecSingle>(c, st); break;
    case 2: stag_pack<PrecHalf>(c, st); break;
  }
}
