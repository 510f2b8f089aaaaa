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
#include "launchers.h"

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

// ---------------------------------------------------------------------------
// zMobius: complex per-slice coefficients b5[s], c5[s] (ref: the zMobius
// branch of dslash_domain_wall_m5.cuh / dslash5_domain_wall.cu — redesigned:
// the HOST assembles, per chirality block, the hop source/weight tables for
// the 5th-dim hop operator and the sequence-ordered bidiagonal (1/diag,
// off-coef, corner) for its inverse; the kernels are generic array-driven
// O(Ls) solvers with no in-kernel index logic. Ls <= 32.)
// ---------------------------------------------------------------------------
template <typename R>
__device__ __forceinline__ cplx<R> zcc(const double c[2]) {
  return {(R)c[0], (R)c[1]};
}

template <typename Prec, bool XPAY>
__global__ __launch_bounds__(256) void k_zdslash5(
    SpinorAcc<Prec> out, SpinorAcc<Prec> in, SpinorAcc<Prec> x, long Vcb4,
    int Ls, typename Prec::Real ar, typename Prec::Real ai,
    const ZCoef *__restrict__ zcp) {
  using R = typename Prec::Real;
  const ZCoef &zc = *zcp;
  long tid = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long n = Vcb4 * Ls;
  if (tid >= n) return;
  long g = tid % Vcb4;
  int s = (int)(tid / Vcb4);
  cplx<R> vin[12], vu[12], vl[12], res[12];
  in.load_v(vin, (long)s * Vcb4 + g);
  in.load_v(vu, (long)zc.su[s] * Vcb4 + g);
  in.load_v(vl, (long)zc.sl[s] * Vcb4 + g);
  cplx<R> au = zcc<R>(zc.au[s]), al = zcc<R>(zc.al[s]);
  cplx<R> wu = zcc<R>(zc.wu[s]), wl = zcc<R>(zc.wl[s]);
#pragma unroll
  for (int k = 0; k < 6; ++k) res[k] = cfma(au, vin[k], wu * vu[k]);
#pragma unroll
  for (int k = 6; k < 12; ++k) res[k] = cfma(al, vin[k], wl * vl[k]);
  if constexpr (XPAY) {
    cplx<R> a{ar, ai}, xv[12];
    x.load_v(xv, (long)s * Vcb4 + g);
#pragma unroll
    for (int k = 0; k < 12; ++k) res[k] = cfma(a, xv[k], res[k]);
  }
  out.store_v(res, (long)s * Vcb4 + g);
}

// Array-driven bidiagonal+corner solve, one thread per 4-d site. Each pass-1
// step does a read-modify-write of ONE chirality half of a slice, so `out`
// must be distinct from `in` (launcher allocates); half precision is NOT
// supported here (the block-float RMW would round the other half) — the
// launcher dispatches double/single only.
template <typename Prec, bool XPAY>
__global__ __launch_bounds__(256) void k_zm5inv(
    SpinorAcc<Prec> out, SpinorAcc<Prec> in, SpinorAcc<Prec> x, long Vcb4,
    int Ls, typename Prec::Real ar, typename Prec::Real ai,
    const ZCoef *__restrict__ zcp) {
  using R = typename Prec::Real;
  const ZCoef &zc = *zcp;
  long g = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (g >= Vcb4) return;
  cplx<R> vin[12], tmp[12];
  cplx<R> prev_u[6], prev_l[6], zu{(R)0, (R)0}, zl{(R)0, (R)0};
  // pass 1: forward substitution in sequence order for both chirality
  // blocks; track z = B^{-1}(corner column) by the same recursion
  for (int i = 0; i < Ls; ++i) {
    int su = zc.ord_u[i], sl = zc.ord_l[i];
    cplx<R> diu = zcc<R>(zc.diu[i]), eu = zcc<R>(zc.eu[i]);
    cplx<R> dil = zcc<R>(zc.dil[i]), el = zcc<R>(zc.el[i]);
    in.load_v(vin, (long)su * Vcb4 + g);
#pragma unroll
    for (int k = 0; k < 6; ++k)
      prev_u[k] = (i == 0) ? diu * vin[k]
                           : diu * (vin[k] - eu * prev_u[k]);
    zu = (i == 0) ? diu * zcc<R>(zc.cwu) : neg(diu * (eu * zu));
    in.load_v(vin, (long)sl * Vcb4 + g);
#pragma unroll
    for (int k = 6; k < 12; ++k)
      prev_l[k - 6] = (i == 0) ? dil * vin[k]
                               : dil * (vin[k] - el * prev_l[k - 6]);
    zl = (i == 0) ? dil * zcc<R>(zc.cwl) : neg(dil * (el * zl));
    // stash y' halves (RMW preserves the other chirality half)
    out.load_v(tmp, (long)su * Vcb4 + g);
#pragma unroll
    for (int k = 0; k < 6; ++k) tmp[k] = prev_u[k];
    out.store_v(tmp, (long)su * Vcb4 + g);
    out.load_v(tmp, (long)sl * Vcb4 + g);
#pragma unroll
    for (int k = 6; k < 12; ++k) tmp[k] = prev_l[k - 6];
    out.store_v(tmp, (long)sl * Vcb4 + g);
  }
  // Sherman-Morrison: y[i] -= z[i] * y'_last / (1 + z_last)
  cplx<R> one{(R)1, (R)0};
  cplx<R> den_u = one + zu, den_l = one + zl;
  cplx<R> fu = ((R)1 / (den_u.re * den_u.re + den_u.im * den_u.im)) * conj(den_u);
  cplx<R> fl = ((R)1 / (den_l.re * den_l.re + den_l.im * den_l.im)) * conj(den_l);
  cplx<R> yu_last[6], yl_last[6];
#pragma unroll
  for (int k = 0; k < 6; ++k) { yu_last[k] = fu * prev_u[k]; yl_last[k] = fl * prev_l[k]; }
  // pass 2: apply corner correction (z recomputed) + optional xpay/scale
  cplx<R> a{ar, ai};
  zu = {(R)0, (R)0};
  zl = {(R)0, (R)0};
  for (int i = 0; i < Ls; ++i) {
    int su = zc.ord_u[i], sl = zc.ord_l[i];
    cplx<R> diu = zcc<R>(zc.diu[i]), dil = zcc<R>(zc.dil[i]);
    zu = (i == 0) ? diu * zcc<R>(zc.cwu) : neg(diu * (zcc<R>(zc.eu[i]) * zu));
    zl = (i == 0) ? dil * zcc<R>(zc.cwl) : neg(dil * (zcc<R>(zc.el[i]) * zl));
    out.load_v(tmp, (long)su * Vcb4 + g);
#pragma unroll
    for (int k = 0; k < 6; ++k) tmp[k] = tmp[k] - zu * yu_last[k];
    out.store_v(tmp, (long)su * Vcb4 + g);
    out.load_v(tmp, (long)sl * Vcb4 + g);
#pragma unroll
    for (int k = 6; k < 12; ++k) tmp[k] = tmp[k] - zl * yl_last[k - 6];
    out.store_v(tmp, (long)sl * Vcb4 + g);
  }
  const bool scale = (ar != (R)1 || ai != (R)0);
  if (XPAY || scale) {
    for (int s = 0; s < Ls; ++s) {
      out.load_v(tmp, (long)s * Vcb4 + g);
      if constexpr (XPAY) {
        cplx<R> xv[12];
        x.load_v(xv, (long)s * Vcb4 + g);
#pragma unroll
        for (int k = 0; k < 12; ++k) tmp[k] = cfma(a, tmp[k], xv[k]);
      } else {
#pragma unroll
        for (int k = 0; k < 12; ++k) tmp[k] = a * tmp[k];
      }
      out.store_v(tmp, (long)s * Vcb4 + g);
    }
  }
}

// ---------------------------------------------------------------------------
// EOFA rank-1 extended M5 ops (ref: kernels/dslash_mobius_eofa.cuh +
// lib/dslash5_mobius_eofa.cu — re-derived: the one-flavor-algorithm 5th-dim
// blocks are the Moebius M5 plus a rank-1 term sh * P_pm |u><w| in s-space;
// the inverse is the plain M5 inverse followed by one Sherman-Morrison
// correction y -= sh * (B^-1 u) (w^dag y) / (1 + sh w^dag B^-1 u), with
// B^-1 u and the denominator precomputed on the HOST. The dagger swaps
// u <-> w (host-side) and daggers the base op.)
// ---------------------------------------------------------------------------

struct EofaVec {  // host-filled (Ls <= 32); real Moebius coefficients
  double u[QA_ZMAX];    // rank-1 left vector (m5_eofa) / B^-1 u (m5inv_eofa)
  double w[QA_ZMAX];    // rank-1 right vector
  double sh;            // shift strength (m5inv: sh/denom folded in)
  int pm;               // +1: rank-1 acts on upper spins (0,1); -1: lower
};

// out(s) = [a*x(s) +] alpha in(s) + beta (Ds in)(s) + sh u_s sum_s' w_s' in_pm(s')
template <typename Prec, bool XPAY, bool DAG>
__global__ __launch_bounds__(256) void k_m5_eofa(
    SpinorAcc<Prec> out, SpinorAcc<Prec> in, SpinorAcc<Prec> x, long Vcb4,
    int Ls, typename Prec::Real a, typename Prec::Real alpha,
    typename Prec::Real beta, typename Prec::Real mf, EofaVec ev) {
  using R = typename Prec::Real;
  long g = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (g >= Vcb4) return;
  cplx<R> vin[12], vu[12], vl[12], res[12];
  // pass A: dot = sum_s w_s in_pm(s)  (one chirality, 6 components)
  cplx<R> dot[6];
#pragma unroll
  for (int k = 0; k < 6; ++k) dot[k] = {(R)0, (R)0};
  const int off = (ev.pm > 0) ? 0 : 6;
  for (int s = 0; s < Ls; ++s) {
    in.load_v(vin, (long)s * Vcb4 + g);
    R ws = (R)ev.w[s];
#pragma unroll
    for (int k = 0; k < 6; ++k) dot[k] = dot[k] + ws * vin[k + off];
  }
  // pass B: base Moebius M5 + rank-1 add
  for (int s = 0; s < Ls; ++s) {
    int su = DAG ? s + 1 : s - 1;
    int sl = DAG ? s - 1 : s + 1;
    R wu = (R)1, wl = (R)1;
    if (su < 0) { su += Ls; wu = -mf; }
    if (su >= Ls) { su -= Ls; wu = -mf; }
    if (sl < 0) { sl += Ls; wl = -mf; }
    if (sl >= Ls) { sl -= Ls; wl = -mf; }
    in.load_v(vin, (long)s * Vcb4 + g);
    in.load_v(vu, (long)su * Vcb4 + g);
    in.load_v(vl, (long)sl * Vcb4 + g);
#pragma unroll
    for (int k = 0; k < 6; ++k) res[k] = alpha * vin[k] + (beta * wu) * vu[k];
#pragma unroll
    for (int k = 6; k < 12; ++k) res[k] = alpha * vin[k] + (beta * wl) * vl[k];
    R c = (R)(ev.sh * ev.u[s]);
#pragma unroll
    for (int k = 0; k < 6; ++k) res[k + off] = res[k + off] + c * dot[k];
    if constexpr (XPAY) {
      cplx<R> xv[12];
      x.load_v(xv, (long)s * Vcb4 + g);
#pragma unroll
      for (int k = 0; k < 12; ++k) res[k] = a * xv[k] + res[k];
    }
    out.store_v(res, (long)s * Vcb4 + g);
  }
}

// out(s) = [x(s) +] a * [(M5 + sh P_pm u w^dag)^{-1} in](s); ev.u holds
// B^-1 u and ev.sh holds sh/denom (host-folded Sherman-Morrison).
template <typename Prec, bool XPAY, bool DAG>
__global__ __launch_bounds__(256) void k_m5inv_eofa(
    SpinorAcc<Prec> out, SpinorAcc<Prec> in, SpinorAcc<Prec> x, long Vcb4,
    int Ls, typename Prec::Real a, typename Prec::Real alpha,
    typename Prec::Real beta, typename Prec::Real mf, EofaVec ev) {
  using R = typename Prec::Real;
  long g = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (g >= Vcb4) return;
  const R kap = -beta / alpha;
  R kpow = (R)1;
  for (int s = 0; s < Ls; ++s) kpow *= kap;
  const R denom = (R)1 + mf * kpow;
  const R inv_alpha = (R)1 / alpha;
  cplx<R> prev_u[6], prev_l[6], vin[12], vout[12];
  // passes 1-3: plain M5 inverse into out (no scale/xpay yet)
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
  R kp = kpow / kap;
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
  kp = kpow / kap;
  for (int si = 0; si < Ls; ++si) {
    int s = DAG ? (Ls - 1 - si) : si;
    out.load_v(vin, (long)s * Vcb4 + g);
#pragma unroll
    for (int k = 6; k < 12; ++k) vin[k] = vin[k] + (cu * kp) * yl_last[k - 6];
    out.store_v(vin, (long)s * Vcb4 + g);
    kp = kp / kap;
  }
  // pass 4: Sherman-Morrison on chirality pm, then scale/xpay epilogue
  const int off = (ev.pm > 0) ? 0 : 6;
  cplx<R> dot[6];
#pragma unroll
  for (int k = 0; k < 6; ++k) dot[k] = {(R)0, (R)0};
  for (int s = 0; s < Ls; ++s) {
    out.load_v(vin, (long)s * Vcb4 + g);
    R ws = (R)ev.w[s];
#pragma unroll
    for (int k = 0; k < 6; ++k) dot[k] = dot[k] + ws * vin[k + off];
  }
  for (int s = 0; s < Ls; ++s) {
    out.load_v(vin, (long)s * Vcb4 + g);
    R c = (R)(ev.sh * ev.u[s]);  // sh/denom * (B^-1 u)_s
#pragma unroll
    for (int k = 0; k < 6; ++k) vin[k + off] = vin[k + off] - c * dot[k];
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
  }
}
