// Native SU(3) heatbath / overrelaxation sweep kernel for MI355X
// (role of reference kernels/gauge_heatbath.cuh + lib/pgauge_heatbath.cu:
//  Cabibbo-Marinari SU(2)-subgroup updates with Kennedy-Pendleton
//  sampling — re-derived; the per-site W = U S^dag is kept IN REGISTERS
//  across the 3 subgroup hits instead of re-materialized per subgroup
//  like the torch fallback).
//
// Operates directly on the oracle complex-double gauge tensor
// [4][2][Vcb][3][3] (thermalization is double-precision Monte Carlo, not
// a bandwidth-critical op: the win over the torch path is launch count —
// 8 launches/sweep instead of ~hundreds of einsum/scatter kernels).
//
// One launch updates U_mu on one checkerboard parity: the staples touch
// only opposite-parity same-direction links and other-direction links,
// so the sweep is race-free by construction.
//
// RNG: counter-based splitmix64 keyed on (seed, site, draw counter) —
// stateless, reproducible for a fixed seed regardless of launch geometry.
#include <hip/hip_runtime.h>

#include "common.h"
#include "launchers.h"

namespace {

struct cd {
  double re, im;
};
__device__ __forceinline__ cd cmul(const cd &a, const cd &b) {
  return {a.re * b.re - a.im * b.im, a.re * b.im + a.im * b.re};
}
__device__ __forceinline__ cd cmulj(const cd &a, const cd &b) {  // a * conj(b)
  return {a.re * b.re + a.im * b.im, a.im * b.re - a.re * b.im};
}
__device__ __forceinline__ cd cjmul(const cd &a, const cd &b) {  // conj(a) * b
  return {a.re * b.re + a.im * b.im, a.re * b.im - a.im * b.re};
}
__device__ __forceinline__ cd cadd(const cd &a, const cd &b) {
  return {a.re + b.re, a.im + b.im};
}

using M3 = cd[3][3];

__device__ __forceinline__ void mat_mul(M3 &o, const M3 &a, const M3 &b) {
#pragma unroll
  for (int i = 0; i < 3; ++i)
#pragma unroll
    for (int j = 0; j < 3; ++j) {
      cd s = cmul(a[i][0], b[0][j]);
      s = cadd(s, cmul(a[i][1], b[1][j]));
      s = cadd(s, cmul(a[i][2], b[2][j]));
      o[i][j] = s;
    }
}

// o = a * b^dag
__device__ __forceinline__ void mat_mul_dag(M3 &o, const M3 &a, const M3 &b) {
#pragma unroll
  for (int i = 0; i < 3; ++i)
#pragma unroll
    for (int j = 0; j < 3; ++j) {
      cd s = cmulj(a[i][0], b[j][0]);
      s = cadd(s, cmulj(a[i][1], b[j][1]));
      s = cadd(s, cmulj(a[i][2], b[j][2]));
      o[i][j] = s;
    }
}

// o = a^dag * b
__device__ __forceinline__ void mat_dag_mul(M3 &o, const M3 &a, const M3 &b) {
#pragma unroll
  for (int i = 0; i < 3; ++i)
#pragma unroll
    for (int j = 0; j < 3; ++j) {
      cd s = cjmul(a[0][i], b[0][j]);
      s = cadd(s, cjmul(a[1][i], b[1][j]));
      s = cadd(s, cjmul(a[2][i], b[2][j]));
      o[i][j] = s;
    }
}

__device__ __forceinline__ void mat_acc(M3 &o, const M3 &a) {
#pragma unroll
  for (int i = 0; i < 3; ++i)
#pragma unroll
    for (int j = 0; j < 3; ++j) o[i][j] = cadd(o[i][j], a[i][j]);
}

// gauge tensor accessor: base + (((mu*2+p)*V + i)*9 + r*3+c)
__device__ __forceinline__ void load_link(M3 &u, const cd *g, long V, int mu,
                                          int p, long i) {
  const cd *b = g + (((long)mu * 2 + p) * V + i) * 9;
#pragma unroll
  for (int k = 0; k < 9; ++k) u[k / 3][k % 3] = b[k];
}

__device__ __forceinline__ void store_link(cd *g, long V, int mu, int p,
                                           long i, const M3 &u) {
  cd *b = g + (((long)mu * 2 + p) * V + i) * 9;
#pragma unroll
  for (int k = 0; k < 9; ++k) b[k] = u[k / 3][k % 3];
}

// splitmix64 -> uniform double in [0,1)
__device__ __forceinline__ unsigned long long sm64(unsigned long long x) {
  x += 0x9e3779b97f4a7c15ULL;
  x = (x ^ (x >> 30)) * 0xbf58476d1ce4e5b9ULL;
  x = (x ^ (x >> 27)) * 0x94d049bb133111ebULL;
  return x ^ (x >> 31);
}

struct Rng {
  unsigned long long key;
  unsigned long long ctr;
  __device__ __forceinline__ double u01() {
    unsigned long long v = sm64(key ^ sm64(ctr++));
    return (double)(v >> 11) * (1.0 / 9007199254740992.0);
  }
};

// Kennedy-Pendleton sample of a0 ~ sqrt(1-a0^2) exp(alpha a0)
__device__ __forceinline__ double kp_a0(double alpha, Rng &rng) {
  for (int it = 0; it < 100; ++it) {
    double r1 = fmax(rng.u01(), 1e-12);
    double r2 = rng.u01();
    double r3 = fmax(rng.u01(), 1e-12);
    double c = cos(2.0 * M_PI * r2);
    double x = -(log(r1) + c * c * log(r3)) / alpha;
    if (x <= 2.0) {
      double a = rng.u01();
      if (a * a <= 1.0 - 0.5 * x) return 1.0 - x;
    }
  }
  return 2.0 * rng.u01() - 1.0;  // pathological alpha fallback
}

}  // namespace

// mode: 0 = heatbath (beta_eff = 2 beta/3), 1 = overrelax (deterministic)
__global__ __launch_bounds__(128) void k_heatbath(
    cd *__restrict__ g, LatDims d, int parity, int mu, double beta_eff,
    unsigned long long seed, int mode) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= d.Vcb) return;
  long V = d.Vcb;
  int xc[4];
  coords_from_cb(xc, i, d, parity);

  // staple sum S (ops.staple_sum convention):
  //   S += U_nu(x) U_mu(x+nu) U_nu(x+mu)^dag
  //   S += U_nu(x-nu)^dag U_mu(x-nu) U_nu(x+mu-nu)
  M3 S;
#pragma unroll
  for (int k = 0; k < 9; ++k) S[k / 3][k % 3] = {0.0, 0.0};
  const int op = 1 - parity;
  long x_pmu = neighbor_cb(xc, mu, +1, d);
#pragma unroll
  for (int nu = 0; nu < 4; ++nu) {
    if (nu == mu) continue;
    M3 a, b, c, t, st;
    long x_pnu = neighbor_cb(xc, nu, +1, d);
    // up staple
    load_link(a, g, V, nu, parity, i);       // U_nu(x)
    load_link(b, g, V, mu, op, x_pnu);       // U_mu(x+nu)
    load_link(c, g, V, nu, op, x_pmu);       // U_nu(x+mu)
    mat_mul(t, a, b);
    mat_mul_dag(st, t, c);
    mat_acc(S, st);
    // down staple
    long x_mnu = neighbor_cb(xc, nu, -1, d);
    int y[4] = {xc[0], xc[1], xc[2], xc[3]};
    y[nu] = y[nu] - 1;
    if (y[nu] < 0) y[nu] += d.X[nu];
    y[mu] = y[mu] + 1;
    if (y[mu] >= d.X[mu]) y[mu] -= d.X[mu];
    long x_pmu_mnu = cb_from_coords(y, d);
    load_link(a, g, V, nu, op, x_mnu);       // U_nu(x-nu)
    load_link(b, g, V, mu, op, x_mnu);       // U_mu(x-nu)
    load_link(c, g, V, nu, parity, x_pmu_mnu);  // U_nu(x+mu-nu)
    mat_dag_mul(t, a, b);
    mat_mul(st, t, c);
    mat_acc(S, st);
  }

  M3 U, W;
  load_link(U, g, V, mu, parity, i);
  mat_mul_dag(W, U, S);  // W = U S^dag

  Rng rng{seed ^ (0x7c15ULL * (unsigned long long)(mu * 2 + parity)),
          (unsigned long long)i * 1024ULL};

  static const int SI[3] = {0, 0, 1};
  static const int SJ[3] = {1, 2, 2};
#pragma unroll
  for (int s = 0; s < 3; ++s) {
    const int si = SI[s], sj = SJ[s];
    // SU(2) projection of the (si,sj) block of W
    double a0 = 0.5 * (W[si][si].re + W[sj][sj].re);
    double a1 = 0.5 * (W[si][sj].im + W[sj][si].im);
    double a2 = 0.5 * (W[si][sj].re - W[sj][si].re);
    double a3 = 0.5 * (W[si][si].im - W[sj][sj].im);
    double k = sqrt(a0 * a0 + a1 * a1 + a2 * a2 + a3 * a3);
    k = fmax(k, 1e-30);
    double v0 = a0 / k, v1 = a1 / k, v2 = a2 / k, v3 = a3 / k;
    double r0, r1, r2, r3;
    if (mode == 0) {
      double b0 = kp_a0(beta_eff * k, rng);
      double rho = sqrt(fmax(1.0 - b0 * b0, 0.0));
      double ct = 2.0 * rng.u01() - 1.0;
      double st = sqrt(fmax(1.0 - ct * ct, 0.0));
      double ph = 2.0 * M_PI * rng.u01();
      double b1 = rho * st * cos(ph);
      double b2 = rho * st * sin(ph);
      double b3 = rho * ct;
      r0 = b0 * v0 + b1 * v1 + b2 * v2 + b3 * v3;
      r1 = -b0 * v1 + b1 * v0 - b2 * v3 + b3 * v2;
      r2 = -b0 * v2 + b2 * v0 - b3 * v1 + b1 * v3;
      r3 = -b0 * v3 + b3 * v0 - b1 * v2 + b2 * v1;
    } else {  // overrelaxation: g = v^-2
      r0 = 2.0 * v0 * v0 - 1.0;
      r1 = -2.0 * v0 * v1;
      r2 = -2.0 * v0 * v2;
      r3 = -2.0 * v0 * v3;
    }
    // left-multiply rows (si,sj) of U and W by the embedded SU(2)
    cd ra = {r0, r3}, rb = {r2, r1};
#pragma unroll
    for (int col = 0; col < 3; ++col) {
      cd ui = U[si][col], uj = U[sj][col];
      U[si][col] = cadd(cmul(ra, ui), cmul(rb, uj));
      U[sj][col] = cadd(cmul(cd{-rb.re, rb.im}, ui), cmulj(uj, ra));
      cd wi = W[si][col], wj = W[sj][col];
      W[si][col] = cadd(cmul(ra, wi), cmul(rb, wj));
      W[sj][col] = cadd(cmul(cd{-rb.re, rb.im}, wi), cmulj(wj, ra));
    }
  }
  store_link(g, V, mu, parity, i, U);
}

void launch_heatbath(const HeatbathCall &c, hipStream_t st) {
  LatDims d{{c.Xdim[0], c.Xdim[1], c.Xdim[2], c.Xdim[3]}, c.parity_offset,
            c.Vcb};
  int grid = (int)((c.Vcb + 127) / 128);
  hipLaunchKernelGGL(k_heatbath, dim3(grid), dim3(128), 0, st, (cd *)c.u, d,
                     c.parity, c.mu, c.beta_eff, c.seed, c.mode);
}

// ---------------------------------------------------------------------------
// Native stout smearing (role of kernels/gauge_stout.cuh):
//   U'_mu = exp(rho * TA[S U_mu^dag]) U_mu
// exp of the traceless-antihermitian Q by scaling-and-squaring with an
// 8-term Taylor series (||rho Q|| <~ 1 for physical rho: scale to < 1/4,
// exact to ~1e-14 — branch-free, unlike the Morningstar-Peardon
// Cayley-Hamilton form with its small-w singularities).
// Reads `gin`, writes `gout` (all links of one mu per launch; every
// staple comes from the OLD field, so the update is order-free).
// ---------------------------------------------------------------------------

namespace {

__device__ __forceinline__ void mat_scale_add_eye(M3 &o, const M3 &a,
                                                  double s) {
#pragma unroll
  for (int i = 0; i < 3; ++i)
#pragma unroll
    for (int j = 0; j < 3; ++j) {
      o[i][j] = {a[i][j].re * s + (i == j ? 1.0 : 0.0), a[i][j].im * s};
    }
}

__device__ __forceinline__ void mat_exp_ta(M3 &E, const M3 &Q) {
  // scale Q by 2^-s so its 1-norm < 0.25
  double n = 0.0;
#pragma unroll
  for (int i = 0; i < 3; ++i)
#pragma unroll
    for (int j = 0; j < 3; ++j)
      n += fabs(Q[i][j].re) + fabs(Q[i][j].im);
  int s = 0;
  double sc = 1.0;
  while (n * sc > 0.25 && s < 40) {
    sc *= 0.5;
    ++s;
  }
  M3 A;
#pragma unroll
  for (int i = 0; i < 3; ++i)
#pragma unroll
    for (int j = 0; j < 3; ++j) A[i][j] = {Q[i][j].re * sc, Q[i][j].im * sc};
  // 8-term Horner Taylor: E = I + A(I + A/2 (I + A/3 (...)))
  M3 T, P;
  mat_scale_add_eye(T, A, 1.0 / 8.0);  // placeholder start: I + A/8
  for (int k = 7; k >= 1; --k) {
    mat_mul(P, A, T);
#pragma unroll
    for (int i = 0; i < 3; ++i)
#pragma unroll
      for (int j = 0; j < 3; ++j) {
        double f = 1.0 / k;
        P[i][j] = {P[i][j].re * f + (i == j ? 1.0 : 0.0), P[i][j].im * f};
      }
#pragma unroll
    for (int i = 0; i < 3; ++i)
#pragma unroll
      for (int j = 0; j < 3; ++j) T[i][j] = P[i][j];
  }
  // square s times
  for (int k = 0; k < s; ++k) {
    mat_mul(P, T, T);
#pragma unroll
    for (int i = 0; i < 3; ++i)
#pragma unroll
      for (int j = 0; j < 3; ++j) T[i][j] = P[i][j];
  }
#pragma unroll
  for (int i = 0; i < 3; ++i)
#pragma unroll
    for (int j = 0; j < 3; ++j) E[i][j] = T[i][j];
}

}  // namespace

__global__ __launch_bounds__(128) void k_stout(
    cd *__restrict__ gout, const cd *__restrict__ gin, LatDims d, int mu,
    double rho) {
  long t = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long V = d.Vcb;
  if (t >= 2 * V) return;
  int parity = (int)(t / V);
  long i = t - (long)parity * V;
  int xc[4];
  coords_from_cb(xc, i, d, parity);

  M3 S;
#pragma unroll
  for (int k = 0; k < 9; ++k) S[k / 3][k % 3] = {0.0, 0.0};
  const int op = 1 - parity;
  long x_pmu = neighbor_cb(xc, mu, +1, d);
#pragma unroll
  for (int nu = 0; nu < 4; ++nu) {
    if (nu == mu) continue;
    M3 a, b, c, tm, st;
    long x_pnu = neighbor_cb(xc, nu, +1, d);
    load_link(a, gin, V, nu, parity, i);
    load_link(b, gin, V, mu, op, x_pnu);
    load_link(c, gin, V, nu, op, x_pmu);
    mat_mul(tm, a, b);
    mat_mul_dag(st, tm, c);
    mat_acc(S, st);
    long x_mnu = neighbor_cb(xc, nu, -1, d);
    int y[4] = {xc[0], xc[1], xc[2], xc[3]};
    y[nu] = y[nu] - 1;
    if (y[nu] < 0) y[nu] += d.X[nu];
    y[mu] = y[mu] + 1;
    if (y[mu] >= d.X[mu]) y[mu] -= d.X[mu];
    long x_pmu_mnu = cb_from_coords(y, d);
    load_link(a, gin, V, nu, op, x_mnu);
    load_link(b, gin, V, mu, op, x_mnu);
    load_link(c, gin, V, nu, parity, x_pmu_mnu);
    mat_dag_mul(tm, a, b);
    mat_mul(st, tm, c);
    mat_acc(S, st);
  }

  M3 U, W, Q, E, Un;
  load_link(U, gin, V, mu, parity, i);
  mat_mul_dag(W, S, U);  // S U^dag
  // Q = rho * TA[W] = rho * ((W - W^dag)/2 - tr/3)
  cd tr = {0.0, 0.0};
#pragma unroll
  for (int i2 = 0; i2 < 3; ++i2)
#pragma unroll
    for (int j = 0; j < 3; ++j) {
      Q[i2][j] = {0.5 * (W[i2][j].re - W[j][i2].re),
                  0.5 * (W[i2][j].im + W[j][i2].im)};
    }
#pragma unroll
  for (int i2 = 0; i2 < 3; ++i2) tr = cadd(tr, Q[i2][i2]);
#pragma unroll
  for (int i2 = 0; i2 < 3; ++i2) {
    Q[i2][i2].re -= tr.re / 3.0;
    Q[i2][i2].im -= tr.im / 3.0;
  }
#pragma unroll
  for (int i2 = 0; i2 < 3; ++i2)
#pragma unroll
    for (int j = 0; j < 3; ++j) {
      Q[i2][j].re *= rho;
      Q[i2][j].im *= rho;
    }
  mat_exp_ta(E, Q);
  mat_mul(Un, E, U);
  store_link(gout, V, mu, parity, i, Un);
}

void launch_stout(const StoutCall &c, hipStream_t st) {
  LatDims d{{c.Xdim[0], c.Xdim[1], c.Xdim[2], c.Xdim[3]}, c.parity_offset,
            c.Vcb};
  int grid = (int)((2 * c.Vcb + 127) / 128);
  hipLaunchKernelGGL(k_stout, dim3(grid), dim3(128), 0, st, (cd *)c.out,
                     (const cd *)c.in, d, c.mu, c.rho);
}

// ---------------------------------------------------------------------------
// Wilson-flow building blocks (role of kernels/gauge_wilson_flow.cuh):
// k_zmat computes Z_mu = eps * TA[S U^dag] into a gauge-shaped tensor;
// k_expmul applies U' = exp(Zc) U for a host-combined Zc (the RK3 stage
// combinations are cheap elementwise torch ops on the Z tensors).
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(128) void k_zmat(
    cd *__restrict__ Z, const cd *__restrict__ gin, LatDims d, int mu,
    double eps) {
  long t = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long V = d.Vcb;
  if (t >= 2 * V) return;
  int parity = (int)(t / V);
  long i = t - (long)parity * V;
  int xc[4];
  coords_from_cb(xc, i, d, parity);

  M3 S;
#pragma unroll
  for (int k = 0; k < 9; ++k) S[k / 3][k % 3] = {0.0, 0.0};
  const int op = 1 - parity;
  long x_pmu = neighbor_cb(xc, mu, +1, d);
#pragma unroll
  for (int nu = 0; nu < 4; ++nu) {
    if (nu == mu) continue;
    M3 a, b, c, tm, st;
    long x_pnu = neighbor_cb(xc, nu, +1, d);
    load_link(a, gin, V, nu, parity, i);
    load_link(b, gin, V, mu, op, x_pnu);
    load_link(c, gin, V, nu, op, x_pmu);
    mat_mul(tm, a, b);
    mat_mul_dag(st, tm, c);
    mat_acc(S, st);
    long x_mnu = neighbor_cb(xc, nu, -1, d);
    int y[4] = {xc[0], xc[1], xc[2], xc[3]};
    y[nu] = y[nu] - 1;
    if (y[nu] < 0) y[nu] += d.X[nu];
    y[mu] = y[mu] + 1;
    if (y[mu] >= d.X[mu]) y[mu] -= d.X[mu];
    long x_pmu_mnu = cb_from_coords(y, d);
    load_link(a, gin, V, nu, op, x_mnu);
    load_link(b, gin, V, mu, op, x_mnu);
    load_link(c, gin, V, nu, parity, x_pmu_mnu);
    mat_dag_mul(tm, a, b);
    mat_mul(st, tm, c);
    mat_acc(S, st);
  }
  M3 U, W, Q;
  load_link(U, gin, V, mu, parity, i);
  mat_mul_dag(W, S, U);
  cd tr = {0.0, 0.0};
#pragma unroll
  for (int i2 = 0; i2 < 3; ++i2)
#pragma unroll
    for (int j = 0; j < 3; ++j)
      Q[i2][j] = {0.5 * (W[i2][j].re - W[j][i2].re),
                  0.5 * (W[i2][j].im + W[j][i2].im)};
#pragma unroll
  for (int i2 = 0; i2 < 3; ++i2) tr = cadd(tr, Q[i2][i2]);
#pragma unroll
  for (int i2 = 0; i2 < 3; ++i2) {
    Q[i2][i2].re -= tr.re / 3.0;
    Q[i2][i2].im -= tr.im / 3.0;
  }
#pragma unroll
  for (int i2 = 0; i2 < 3; ++i2)
#pragma unroll
    for (int j = 0; j < 3; ++j) {
      Q[i2][j].re *= eps;
      Q[i2][j].im *= eps;
    }
  store_link(Z, V, mu, parity, i, Q);
}

__global__ __launch_bounds__(128) void k_expmul(
    cd *__restrict__ gout, const cd *__restrict__ gin,
    const cd *__restrict__ Zc, LatDims d, int mu) {
  long t = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long V = d.Vcb;
  if (t >= 2 * V) return;
  int parity = (int)(t / V);
  long i = t - (long)parity * V;
  M3 Q, E, U, Un;
  load_link(Q, Zc, V, mu, parity, i);
  load_link(U, gin, V, mu, parity, i);
  mat_exp_ta(E, Q);
  mat_mul(Un, E, U);
  store_link(gout, V, mu, parity, i, Un);
}

void launch_zmat(const StoutCall &c, hipStream_t st) {
  LatDims d{{c.Xdim[0], c.Xdim[1], c.Xdim[2], c.Xdim[3]}, c.parity_offset,
            c.Vcb};
  int grid = (int)((2 * c.Vcb + 127) / 128);
  hipLaunchKernelGGL(k_zmat, dim3(grid), dim3(128), 0, st, (cd *)c.out,
                     (const cd *)c.in, d, c.mu, c.rho);
}

void launch_expmul(const StoutCall &c, hipStream_t st) {
  LatDims d{{c.Xdim[0], c.Xdim[1], c.Xdim[2], c.Xdim[3]}, c.parity_offset,
            c.Vcb};
  int grid = (int)((2 * c.Vcb + 127) / 128);
  hipLaunchKernelGGL(k_expmul, dim3(grid), dim3(128), 0, st, (cd *)c.out,
                     (const cd *)c.in, (const cd *)c.aux, d, c.mu);
}
