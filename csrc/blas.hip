// Fused BLAS + reduction kernels (role of reference lib/blas_quda.cu /
// lib/reduce_quda.cu functors, kernels/blas_core.cuh + reduce_core.cuh).
// Site-structured so HALF (per-site norm) precision composes with every op;
// arithmetic always in Real (float for half), reductions accumulate double.
// Block 256 (4 waves), grid-stride; block-reduce -> one f64 atomic per block.
#include "common.h"

// ---------------------------------------------------------------------------
// reduction plumbing
// ---------------------------------------------------------------------------
__device__ __forceinline__ double wave_reduce(double v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return v;
}

__device__ __forceinline__ void block_atomic_add(double v, double *out) {
  __shared__ double partial[4];
  int lane = threadIdx.x & 63;
  int wave = threadIdx.x >> 6;
  v = wave_reduce(v);
  if (lane == 0) partial[wave] = v;
  __syncthreads();
  if (wave == 0) {
    double s = (lane < (int)(blockDim.x >> 6)) ? partial[lane] : 0.0;
    s = wave_reduce(s);
    if (lane == 0) atomicAdd(out, s);
  }
}

// ---------------------------------------------------------------------------
// site-wise blas: each functor consumes/produces [4][3] complex site values
// ---------------------------------------------------------------------------
template <typename Prec>
using Site = cplx<typename Prec::Real>[4][3];

#define GRID_STRIDE(g, n)                                                     \
  for (long g = (long)blockIdx.x * blockDim.x + threadIdx.x; g < (n);         \
       g += (long)gridDim.x * blockDim.x)

// y = a*x + y ; optional norm2(y) accumulation
template <typename Prec, bool NORM2>
__global__ __launch_bounds__(256) void k_axpy(
    typename Prec::Real a, SpinorAcc<Prec> x, SpinorAcc<Prec> y, long sites,
    double *result) {
  using R = typename Prec::Real;
  double acc = 0.0;
  GRID_STRIDE(g, sites) {
    cplx<R> xv[4][3], yv[4][3];
    x.load_g(xv, g);
    y.load_g(yv, g);
#pragma unroll
    for (int s = 0; s < 4; ++s)
#pragma unroll
      for (int c = 0; c < 3; ++c) {
        yv[s][c] = yv[s][c] + a * xv[s][c];
        if constexpr (NORM2)
          acc += (double)yv[s][c].re * yv[s][c].re + (double)yv[s][c].im * yv[s][c].im;
      }
    y.store_g(yv, g);
  }
  if constexpr (NORM2) block_atomic_add(acc, result);
}

// y = x + a*y
template <typename Prec>
__global__ __launch_bounds__(256) void k_xpay(
    SpinorAcc<Prec> x, typename Prec::Real a, SpinorAcc<Prec> y, long sites) {
  using R = typename Prec::Real;
  GRID_STRIDE(g, sites) {
    cplx<R> xv[4][3], yv[4][3];
    x.load_g(xv, g);
    y.load_g(yv, g);
#pragma unroll
    for (int s = 0; s < 4; ++s)
#pragma unroll
      for (int c = 0; c < 3; ++c) yv[s][c] = xv[s][c] + a * yv[s][c];
    y.store_g(yv, g);
  }
}

// y = a*x + b*y
template <typename Prec>
__global__ __launch_bounds__(256) void k_axpby(
    typename Prec::Real a, SpinorAcc<Prec> x, typename Prec::Real b,
    SpinorAcc<Prec> y, long sites) {
  using R = typename Prec::Real;
  GRID_STRIDE(g, sites) {
    cplx<R> xv[4][3], yv[4][3];
    x.load_g(xv, g);
    y.load_g(yv, g);
#pragma unroll
    for (int s = 0; s < 4; ++s)
#pragma unroll
      for (int c = 0; c < 3; ++c) yv[s][c] = a * xv[s][c] + b * yv[s][c];
    y.store_g(yv, g);
  }
}

// y += (ar + i ai) * x
template <typename Prec>
__global__ __launch_bounds__(256) void k_caxpy(
    typename Prec::Real ar, typename Prec::Real ai, SpinorAcc<Prec> x,
    SpinorAcc<Prec> y, long sites) {
  using R = typename Prec::Real;
  cplx<R> a{ar, ai};
  GRID_STRIDE(g, sites) {
    cplx<R> xv[4][3], yv[4][3];
    x.load_g(xv, g);
    y.load_g(yv, g);
#pragma unroll
    for (int s = 0; s < 4; ++s)
#pragma unroll
      for (int c = 0; c < 3; ++c) yv[s][c] = cfma(a, xv[s][c], yv[s][c]);
    y.store_g(yv, g);
  }
}

// y = (ar+i ai)*x + (br+i bi)*y
template <typename Prec>
__global__ __launch_bounds__(256) void k_caxpby(
    typename Prec::Real ar, typename Prec::Real ai, SpinorAcc<Prec> x,
    typename Prec::Real br, typename Prec::Real bi, SpinorAcc<Prec> y,
    long sites) {
  using R = typename Prec::Real;
  cplx<R> a{ar, ai}, b{br, bi};
  GRID_STRIDE(g, sites) {
    cplx<R> xv[4][3], yv[4][3];
    x.load_g(xv, g);
    y.load_g(yv, g);
#pragma unroll
    for (int s = 0; s < 4; ++s)
#pragma unroll
      for (int c = 0; c < 3; ++c) yv[s][c] = cfma(a, xv[s][c], b * yv[s][c]);
    y.store_g(yv, g);
  }
}

// y = x - y ; returns norm2(y)
template <typename Prec>
__global__ __launch_bounds__(256) void k_xmy_norm2(
    SpinorAcc<Prec> x, SpinorAcc<Prec> y, long sites, double *result) {
  using R = typename Prec::Real;
  double acc = 0.0;
  GRID_STRIDE(g, sites) {
    cplx<R> xv[4][3], yv[4][3];
    x.load_g(xv, g);
    y.load_g(yv, g);
#pragma unroll
    for (int s = 0; s < 4; ++s)
#pragma unroll
      for (int c = 0; c < 3; ++c) {
        yv[s][c] = xv[s][c] - yv[s][c];
        acc += (double)yv[s][c].re * yv[s][c].re + (double)yv[s][c].im * yv[s][c].im;
      }
    y.store_g(yv, g);
  }
  block_atomic_add(acc, result);
}

// x *= a
template <typename Prec>
__global__ __launch_bounds__(256) void k_scal(
    typename Prec::Real a, SpinorAcc<Prec> x, long sites) {
  using R = typename Prec::Real;
  GRID_STRIDE(g, sites) {
    cplx<R> xv[4][3];
    x.load_g(xv, g);
#pragma unroll
    for (int s = 0; s < 4; ++s)
#pragma unroll
      for (int c = 0; c < 3; ++c) xv[s][c] = a * xv[s][c];
    x.store_g(xv, g);
  }
}

// reductions: norm2, re<x,y>, <x,y> (re+im)
template <typename Prec, int KIND>  // 0 norm2, 1 redot, 2 cdot
__global__ __launch_bounds__(256) void k_reduce(
    SpinorAcc<Prec> x, SpinorAcc<Prec> y, long sites, double *result) {
  using R = typename Prec::Real;
  double acc = 0.0, acc2 = 0.0;
  GRID_STRIDE(g, sites) {
    cplx<R> xv[4][3], yv[4][3];
    x.load_g(xv, g);
    if constexpr (KIND != 0) y.load_g(yv, g);
#pragma unroll
    for (int s = 0; s < 4; ++s)
#pragma unroll
      for (int c = 0; c < 3; ++c) {
        if constexpr (KIND == 0) {
          acc += (double)xv[s][c].re * xv[s][c].re + (double)xv[s][c].im * xv[s][c].im;
        } else if constexpr (KIND == 1) {
          acc += (double)xv[s][c].re * yv[s][c].re + (double)xv[s][c].im * yv[s][c].im;
        } else {
          acc += (double)xv[s][c].re * yv[s][c].re + (double)xv[s][c].im * yv[s][c].im;
          acc2 += (double)xv[s][c].re * yv[s][c].im - (double)xv[s][c].im * yv[s][c].re;
        }
      }
  }
  block_atomic_add(acc, result);
  if constexpr (KIND == 2) block_atomic_add(acc2, result + 1);
}

// precision conversion copy (ref: lib/copy_color_spinor_*.cu)
template <typename PrecDst, typename PrecSrc>
__global__ __launch_bounds__(256) void k_convert(
    SpinorAcc<PrecDst> dst, SpinorAcc<PrecSrc> src, long sites) {
  using RS = typename PrecSrc::Real;
  using RD = typename PrecDst::Real;
  GRID_STRIDE(g, sites) {
    cplx<RS> v[4][3];
    src.load_g(v, g);
    cplx<RD> o[4][3];
#pragma unroll
    for (int s = 0; s < 4; ++s)
#pragma unroll
      for (int c = 0; c < 3; ++c) o[s][c] = {(RD)v[s][c].re, (RD)v[s][c].im};
    dst.store_g(o, g);
  }
}

// ---------------------------------------------------------------------------
// launchers (C ABI consumed by bindings.cpp)
// ---------------------------------------------------------------------------
#include "launchers.h"

namespace {
constexpr int BLK = 256;
int grid_for(long sites) {
  long g = (sites + BLK - 1) / BLK;
  // cap + grid-stride (cdna_hip_programming.md Guideline 11)
  return (int)(g < 2048 ? g : 2048);
}

template <typename Prec>
SpinorAcc<Prec> acc_of(const BlasField &f) {
  return SpinorAcc<Prec>{(typename Prec::Store *)f.data, (float *)f.norm, f.Vcb};
}
}  // namespace

template <typename Prec>
static void blas_dispatch(const BlasCall &c, hipStream_t st) {
  auto x = acc_of<Prec>(c.x);
  auto y = acc_of<Prec>(c.y);
  long n = c.sites;
  int gr = grid_for(n);
  using R = typename Prec::Real;
  switch (c.op) {
    case BLAS_AXPY:
      hipLaunchKernelGGL((k_axpy<Prec, false>), dim3(gr), dim3(BLK), 0, st,
                         (R)c.a, x, y, n, nullptr);
      break;
    case BLAS_AXPY_NORM2:
      hipLaunchKernelGGL((k_axpy<Prec, true>), dim3(gr), dim3(BLK), 0, st,
                         (R)c.a, x, y, n, c.result);
      break;
    case BLAS_XPAY:
      hipLaunchKernelGGL((k_xpay<Prec>), dim3(gr), dim3(BLK), 0, st, x, (R)c.a, y, n);
      break;
    case BLAS_AXPBY:
      hipLaunchKernelGGL((k_axpby<Prec>), dim3(gr), dim3(BLK), 0, st, (R)c.a, x,
                         (R)c.b, y, n);
      break;
    case BLAS_CAXPY:
      hipLaunchKernelGGL((k_caxpy<Prec>), dim3(gr), dim3(BLK), 0, st, (R)c.a,
                         (R)c.b, x, y, n);
      break;
    case BLAS_CAXPBY:
      hipLaunchKernelGGL((k_caxpby<Prec>), dim3(gr), dim3(BLK), 0, st, (R)c.a,
                         (R)c.b, x, (R)c.c, (R)c.d, y, n);
      break;
    case BLAS_XMY_NORM2:
      hipLaunchKernelGGL((k_xmy_norm2<Prec>), dim3(gr), dim3(BLK), 0, st, x, y, n,
                         c.result);
      break;
    case BLAS_SCAL:
      hipLaunchKernelGGL((k_scal<Prec>), dim3(gr), dim3(BLK), 0, st, (R)c.a, x, n);
      break;
    case BLAS_NORM2:
      hipLaunchKernelGGL((k_reduce<Prec, 0>), dim3(gr), dim3(BLK), 0, st, x, y, n,
                         c.result);
      break;
    case BLAS_REDOT:
      hipLaunchKernelGGL((k_reduce<Prec, 1>), dim3(gr), dim3(BLK), 0, st, x, y, n,
                         c.result);
      break;
    case BLAS_CDOT:
      hipLaunchKernelGGL((k_reduce<Prec, 2>), dim3(gr), dim3(BLK), 0, st, x, y, n,
                         c.result);
      break;
  }
}

void launch_blas(const BlasCall &c, hipStream_t st) {
  switch (c.prec) {
    case 0: blas_dispatch<PrecDouble>(c, st); break;
    case 1: blas_dispatch<PrecSingle>(c, st); break;
    case 2: blas_dispatch<PrecHalf>(c, st); break;
  }
}

template <typename PD, typename PS>
static void conv(const BlasField &d, const BlasField &s, long sites, hipStream_t st) {
  auto da = SpinorAcc<PD>{(typename PD::Store *)d.data, (float *)d.norm, d.Vcb};
  auto sa = SpinorAcc<PS>{(typename PS::Store *)s.data, (float *)s.norm, s.Vcb};
  hipLaunchKernelGGL((k_convert<PD, PS>), dim3(grid_for(sites)), dim3(BLK), 0, st,
                     da, sa, sites);
}

void launch_convert(const BlasField &dst, int pdst, const BlasField &src,
                    int psrc, long sites, hipStream_t st) {
  switch (pdst * 3 + psrc) {
    case 0 * 3 + 1: conv<PrecDouble, PrecSingle>(dst, src, sites, st); break;
    case 0 * 3 + 2: conv<PrecDouble, PrecHalf>(dst, src, sites, st); break;
    case 1 * 3 + 0: conv<PrecSingle, PrecDouble>(dst, src, sites, st); break;
    case 1 * 3 + 2: conv<PrecSingle, PrecHalf>(dst, src, sites, st); break;
    case 2 * 3 + 0: conv<PrecHalf, PrecDouble>(dst, src, sites, st); break;
    case 2 * 3 + 1: conv<PrecHalf, PrecSingle>(dst, src, sites, st); break;
    case 0 * 3 + 0: conv<PrecDouble, PrecDouble>(dst, src, sites, st); break;
    case 1 * 3 + 1: conv<PrecSingle, PrecSingle>(dst, src, sites, st); break;
    case 2 * 3 + 2: conv<PrecHalf, PrecHalf>(dst, src, sites, st); break;
  }
}
