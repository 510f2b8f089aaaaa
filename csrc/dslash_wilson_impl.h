// Shared runtime->template dispatch for the Wilson dslash TUs.
#pragma once
#include "dslash_wilson.h"
#include "launchers.h"

template <typename Prec, int RECON>
static void dslash_launch_all(const DslashCall &c, hipStream_t st) {
  using R = typename Prec::Real;
  SpinorAcc<Prec> out{(typename Prec::Store *)c.out.data, (float *)c.out.norm, c.Vcb};
  SpinorAcc<Prec> in{(typename Prec::Store *)c.in.data, (float *)c.in.norm, c.Vcb};
  SpinorAcc<Prec> x{(typename Prec::Store *)c.x.data, (float *)c.x.norm, c.Vcb};
  // parity-offset the stencil gauge base: [2][NCH][V][W]
  const auto *gbase = (const typename Prec::Store *)c.gauge +
                      (long)c.parity * GaugeAcc<Prec, RECON>::NCH * c.Vcb * Prec::W;
  GaugeAcc<Prec, RECON> g{gbase, c.Vcb};
  CloverAcc<Prec> cl{(const typename Prec::Store *)c.clover, c.Vcb};
  LatDims d{{c.Xdim[0], c.Xdim[1], c.Xdim[2], c.Xdim[3]}, c.parity_offset, c.Vcb};
  int blk = 256;
  int grid = (int)((c.Vcb + blk - 1) / blk);
  R a = (R)c.a;

#define QA_LAUNCH(DAG, MODE, XPAY)                                            \
  hipLaunchKernelGGL((k_dslash_wilson<Prec, RECON, DAG, MODE, XPAY>),         \
                     dim3(grid), dim3(blk), 0, st, out, in, g, cl, d,         \
                     c.parity, a, x)

  if (!c.dagger) {
    if (c.mode == PLAIN && !c.xpay) QA_LAUNCH(false, PLAIN, false);
    else if (c.mode == PLAIN) QA_LAUNCH(false, PLAIN, true);
    else if (c.mode == CLOV_POST && !c.xpay) QA_LAUNCH(false, CLOV_POST, false);
    else if (c.mode == CLOV_POST) QA_LAUNCH(false, CLOV_POST, true);
    else QA_LAUNCH(false, CLOV_X, true);
  } else {
    if (c.mode == PLAIN && !c.xpay) QA_LAUNCH(true, PLAIN, false);
    else if (c.mode == PLAIN) QA_LAUNCH(true, PLAIN, true);
    else if (c.mode == CLOV_POST && !c.xpay) QA_LAUNCH(true, CLOV_POST, false);
    else if (c.mode == CLOV_POST) QA_LAUNCH(true, CLOV_POST, true);
    else QA_LAUNCH(true, CLOV_X, true);
  }
#undef QA_LAUNCH
}
