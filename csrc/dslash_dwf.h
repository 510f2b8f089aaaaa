// Domain-wall / Moebius 5th-dimension kernels for MI355X
// (role of reference include/kernels/dslash_domain_wall_m5.cuh — redesigned:
//  in DeGrand-Rossi the chiral projectors P± = (1±g5)/2 are spin-diagonal,
//  so the s-hops never mix spin components; the M5 inverse is a pair of
//  bidiagonal+corner solves done in O(Ls) per site with a Sherman-Morrison
//  corner correction, not a shared-memory scan).
//
// 5-d layout: the 5-d field is ONE chunked field with site index
// i5 = s*Vcb4 + x (chunk stride Ls*Vcb4); 4-d slices are pointer offsets.
//
// s-structure (fermion mass mf, Ls slices):
//   (Ds psi)(s) = P+ psi(s-1) + P- psi(s+1)
//   corners: s=0 upper gets -mf psi_u(Ls-1); s=Ls-1 lower gets -mf psi_l(0)
//   (P+ = upper spins 0,1 in DeGrand-Rossi, P- = lower spins 2,3)
// DAG swaps the hop directions (Ds^dag).
//
// k_dslash5:  out(s) = [a*x(s) +] alpha*in(s) + beta*(Ds in)(s)
// k_m5inv  :  out(s) = [x(s) +] a * [(alpha + beta Ds)^{-1} in](s)
#pragma once

#include "common.h"

template <typename Prec, bool XPAY, bool DAG>
__global__ __launch_bounds__(256) void k_dslash5(
    SpinorAcc<Prec> out, SpinorAcc<Prec> in, SpinorAcc<Prec> x, long Vcb4,
    int Ls, typename Prec::Real a, typename Prec::Real alpha,
    typename Prec::Real beta, typename Prec::Real mf) {
  using R = typename Prec::Real;
  long tid = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long n = Vcb4 * Ls;
  if (tid >= n) return;
  long g = tid % Vcb4;
  int s = (int)(tid / Vcb4);

  // hop source slices for upper/lower blocks
  int su = DAG ? s + 1 : s - 1;  // upper block source
  int sl = DAG ? s - 1 : s + 1;  // lower block source
  R wu = (R)1, wl = (R)1;
  if (su < 0) { su += Ls; wu = -mf; }
  if (su >= Ls) { su -= Ls; wu = -mf; }
  if (sl < 0) { sl += Ls; wl = -mf; }
  if (sl >= Ls) { sl -= Ls; wl = -mf; }

  cplx<R> vin[12], vu[12], vl[12], res[12];
  in.load_v(vin, (long)s * Vcb4 + g);
  in.load_v(vu, (long)su * Vcb4 + g);
  in.load_v(vl, (long)sl * Vcb4 + g);
#pragma unroll
  for (int k = 0; k < 6; ++k) {  // upper block: components 0..5
    res[k] = alpha * vin[k] + (beta * wu) * vu[k];
  }
#pragma unroll
  for (int k = 6; k < 12; ++k) {  // lower block
    res[k] = alpha * vin[k] + (beta * wl) * vl[k];
  }
  if constexpr (XPAY) {
    cplx<R> xv[12];
    x.load_v(xv, (long)s * Vcb4 + g);
#pragma unroll
    for (int k = 0; k < 12; ++k) res[k] = a * xv[k] + res[k];
  }
  out.store_v(res, (long)s * Vcb4 + g);
}

// M5 inverse: thread per 4-d site; three sweeps over s (forward upper
// recursion, backward lower recursion + upper corner fix, forward lower
// corner fix). kap = -beta/alpha; denom = 1 + mf kap^Ls.
template <typename Prec, bool XPAY, bool DAG>
__global__ __launch_bounds__(256) void k_m5inv(
    SpinorAcc<Prec> out, SpinorAcc<Prec> in, SpinorAcc<Prec> x, long Vcb4,
    int Ls, typename Prec::Real a, typename Prec::Real alpha,
    typename Prec::Real beta, typename Prec::Real mf) {
  using R = typename Prec::Real;
  long g = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (g >= Vcb4) return;
  const R kap = -beta / alpha;
  R kpow = (R)1;  // kap^Ls
  for (int s = 0; s < Ls; ++s) kpow *= kap;
  const R denom = (R)1 + mf * kpow;
  const R inv_alpha = (R)1 / alpha;

  // With DAG the upper block recursion runs backward instead of forward
  // (hop directions swap); "fwd" below means ascending s for the upper
  // block in the non-dagger case.
  cplx<R> prev_u[6], prev_l[6], vin[12], vout[12];

  // pass 1 (ascending): y'_u forward recursion; stash raw lower
  for (int si = 0; si < Ls; ++si) {
    int s = DAG ? (Ls - 1 - si) : si;
    in.load_v(vin, (long)s * Vcb4 + g);
#pragma unroll
    for (int k = 0; k < 6; ++k) {
      cplx<R> yu = (si == 0) ? inv_alpha * vin[k]
                             : inv_alpha * (vin[k] - beta * prev_u[k]);
      prev_u[k] = yu;
      vout[k] = yu;
    }
#pragma unroll
    for (int k = 6; k < 12; ++k) vout[k] = vin[k];
    out.store_v(vout, (long)s * Vcb4 + g);
  }
  cplx<R> yu_last[6];
#pragma unroll
  for (int k = 0; k < 6; ++k) yu_last[k] = prev_u[k];

  // pass 2 (descending): y'_l recursion + upper corner correction
  R kp = kpow / kap;  // kap^(Ls-1) -> runs down to kap^0
  const R cu = mf * beta * inv_alpha / denom;
  for (int si = 0; si < Ls; ++si) {
    int s = DAG ? si : (Ls - 1 - si);
    out.load_v(vin, (long)s * Vcb4 + g);
#pragma unroll
    for (int k = 0; k < 6; ++k) vin[k] = vin[k] + (cu * kp) * yu_last[k];
#pragma unroll
    for (int k = 6; k < 12; ++k) {
      cplx<R> yl = (si == 0) ? inv_alpha * vin[k]
                             : inv_alpha * (vin[k] - beta * prev_l[k - 6]);
      prev_l[k - 6] = yl;
      vin[k] = yl;
    }
    out.store_v(vin, (long)s * Vcb4 + g);
    kp = kp / kap;
  }
  cplx<R> yl_last[6];
#pragma unroll
  for (int k = 0; k < 6; ++k) yl_last[k] = prev_l[k];

  // pass 3 (ascending): lower corner correction (+ optional xpay + scale)
  kp = kpow / kap;
  for (int si = 0; si < Ls; ++si) {
    int s = DAG ? (Ls - 1 - si) : si;
    out.load_v(vin, (long)s * Vcb4 + g);
#pragma unroll
    for (int k = 6; k < 12; ++k) vin[k] = vin[k] + (cu * kp) * yl_last[k - 6];
    if constexpr (XPAY) {
      cplx<R> xv[12];
      x.load_v(xv, (long)s * Vcb4 + g);
#pragma unroll
      for (int k = 0; k < 12; ++k) vin[k] = xv[k] + a * vin[k];
    } else {
      if (a != (R)1) {
#pragma unroll
        for (int k = 0; k < 12; ++k) vin[k] = a * vin[k];
      }
    }
    out.store_v(vin, (long)s * Vcb4 + g);
    kp = kp / kap;
  }
}
