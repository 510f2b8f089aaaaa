// Multi-RHS Wilson(-clover) dslash: NRHS right-hand sides per gauge load.
// (role of the reference's MAX_MULTI_RHS kernel batching,
//  dslash_wilson.cuh:38-40 — redesigned direction-major for CDNA4: the
//  single-RHS kernel is load-ISSUE-bound at ~13 B/cyc/CU, so amortizing
//  the gauge (8 links) and clover (144 B) streams over NRHS sides cuts
//  both issued instructions and unique HBM bytes per RHS.)
//
// Loop structure: for each of the 8 hops, load U once, then gather and
// accumulate all NRHS neighbor spinors under it (the NRHS loads are
// independent -> per-direction ILP replaces the single-RHS kernel's
// 8-direction gather phase). Modes: PLAIN and CLOV_POST only (the CG
// family's hot ops); other epilogues take the per-RHS kernel.
#pragma once

#include "common.h"
#include "dslash_wilson.h"
#include "generated/proj.h"
#include "halo.h"

template <typename Prec, int NRHS>
struct MrhsPtrs {
  typename Prec::Store *out[NRHS];
  float *out_n[NRHS];
  const typename Prec::Store *in[NRHS];
  const float *in_n[NRHS];
  const typename Prec::Store *x[NRHS];
  const float *x_n[NRHS];
};

template <typename Prec, int RECON, bool DAG, int MODE, bool XPAY, int KT,
          int NRHS>
__global__ __launch_bounds__(256) void k_dslash_wilson_mrhs(
    MrhsPtrs<Prec, NRHS> ptr, long Vs, GaugeAcc<Prec, RECON> g,
    CloverAcc<Prec> clov, LatDims d, int parity, typename Prec::Real a,
    GhostAcc<Prec> gh) {
  using R = typename Prec::Real;
  static_assert(MODE == PLAIN || MODE == CLOV_POST);
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= d.Vcb) return;
  int xc[4];
  coords_from_cb(xc, i, d, parity);

  cplx<R> acc[NRHS][4][3];
#pragma unroll
  for (int r = 0; r < NRHS; ++r)
#pragma unroll
    for (int s = 0; s < 4; ++s)
#pragma unroll
      for (int c = 0; c < 3; ++c) acc[r][s][c] = {(R)0, (R)0};

  cplx<R> uh[2][3], U[3][3];
  // all NRHS projected half-spinors of a hop live simultaneously so their
  // NRHS x 4 chunk loads issue back-to-back — the gather-first discipline
  // of the single-RHS kernel (profiles/r01_dslash_pmc_analysis.md); a
  // shared scratch buffer here serializes the loads and measures 2-4x
  // slower (r02 finding).
  cplx<R> hh[NRHS][2][3];
  const R one = (R)0.5;

  bool bnd = false;
  if constexpr (KT == KT_INTERIOR) {
#pragma unroll
    for (int m = 0; m < 4; ++m)
      if (gh.active(m) && (xc[m] == 0 || xc[m] == d.X[m] - 1)) bnd = true;
  }

  SpinorAcc<Prec> in_r;  // per-RHS view, pointer swapped in the loop
  in_r.V = Vs;

  // one direction at a time: U loaded once, NRHS gathers under it
#define QA_MDIR(MU)                                                        \
  {                                                                        \
    bool cross_p = KT != KT_LOCAL && gh.active(MU) && xc[MU] == d.X[MU] - 1; \
    if (!(KT == KT_INTERIOR && cross_p)) {                                 \
      g.template load<MU>(U, i);                                           \
      long j = neighbor_cb(xc, MU, +1, d);                                 \
      _Pragma("unroll") for (int r = 0; r < NRHS; ++r) {                   \
        if (KT == KT_FUSED && cross_p) {                                   \
          gh.load_r(hh[r], MU, 1, ghost_idx(xc, MU, d), r);                \
        } else {                                                           \
          cplx<R> p[4][3];                                                 \
          in_r.data = const_cast<typename Prec::Store *>(ptr.in[r]);       \
          in_r.norm = const_cast<float *>(ptr.in_n[r]);                    \
          in_r.load(p, j);                                                 \
          if constexpr (!DAG) proj_##MU##_0(hh[r], p);                     \
          else proj_##MU##_1(hh[r], p);                                    \
        }                                                                  \
      }                                                                    \
      _Pragma("unroll") for (int r = 0; r < NRHS; ++r) {                   \
        su3_mul_half(uh, U, hh[r]);                                        \
        if constexpr (!DAG) recon_##MU##_0(acc[r], uh, one);               \
        else recon_##MU##_1(acc[r], uh, one);                              \
      }                                                                    \
    }                                                                      \
    bool cross_m = KT != KT_LOCAL && gh.active(MU) && xc[MU] == 0;         \
    if (!(KT == KT_INTERIOR && cross_m)) {                                 \
      long j = -1;                                                         \
      if (KT == KT_FUSED && cross_m) {                                     \
        g.template load<4 + MU>(U, i);                                     \
      } else {                                                             \
        j = neighbor_cb(xc, MU, -1, d);                                    \
        g.template load_o<MU>(U, j);                                       \
      }                                                                    \
      _Pragma("unroll") for (int r = 0; r < NRHS; ++r) {                   \
        if (KT == KT_FUSED && cross_m) {                                   \
          gh.load_r(hh[r], MU, 0, ghost_idx(xc, MU, d), r);                \
        } else {                                                           \
          cplx<R> p[4][3];                                                 \
          in_r.data = const_cast<typename Prec::Store *>(ptr.in[r]);       \
          in_r.norm = const_cast<float *>(ptr.in_n[r]);                    \
          in_r.load(p, j);                                                 \
          if constexpr (!DAG) proj_##MU##_1(hh[r], p);                     \
          else proj_##MU##_0(hh[r], p);                                    \
        }                                                                  \
      }                                                                    \
      _Pragma("unroll") for (int r = 0; r < NRHS; ++r) {                   \
        su3_dagmul_half(uh, U, hh[r]);                                     \
        if constexpr (!DAG) recon_##MU##_1(acc[r], uh, one);               \
        else recon_##MU##_0(acc[r], uh, one);                              \
      }                                                                    \
    }                                                                      \
  }

  QA_MDIR(0)
  QA_MDIR(1)
  QA_MDIR(2)
  QA_MDIR(3)
#undef QA_MDIR

  SpinorAcc<Prec> out_r;
  out_r.V = Vs;

  // boundary sites under CLOV_POST defer the whole epilogue (exterior
  // kernel completes it per RHS)
  if constexpr (KT == KT_INTERIOR && MODE == CLOV_POST) {
    if (bnd) {
#pragma unroll
      for (int r = 0; r < NRHS; ++r) {
        out_r.data = ptr.out[r];
        out_r.norm = ptr.out_n[r];
        out_r.store(acc[r], i);
      }
      return;
    }
  }

  R diag[2][6];
  cplx<R> tri[2][15];
  if constexpr (MODE == CLOV_POST) clov.load(diag, tri, parity, i);

  SpinorAcc<Prec> x_r;
  x_r.V = Vs;
#pragma unroll
  for (int r = 0; r < NRHS; ++r) {
    if constexpr (MODE == CLOV_POST) {
      cplx<R> tmp[4][3];
      clover_mul(tmp, diag, tri, acc[r]);
#pragma unroll
      for (int s = 0; s < 4; ++s)
#pragma unroll
        for (int c = 0; c < 3; ++c) acc[r][s][c] = tmp[s][c];
    }
    if constexpr (XPAY) {
      cplx<R> xv[4][3];
      x_r.data = const_cast<typename Prec::Store *>(ptr.x[r]);
      x_r.norm = const_cast<float *>(ptr.x_n[r]);
      x_r.load(xv, i);
#pragma unroll
      for (int s = 0; s < 4; ++s)
#pragma unroll
        for (int c = 0; c < 3; ++c) acc[r][s][c] = xv[s][c] + a * acc[r][s][c];
    } else if (a != (R)1) {
#pragma unroll
      for (int s = 0; s < 4; ++s)
#pragma unroll
        for (int c = 0; c < 3; ++c) acc[r][s][c] = a * acc[r][s][c];
    }
    out_r.data = ptr.out[r];
    out_r.norm = ptr.out_n[r];
    out_r.store(acc[r], i);
  }
}
