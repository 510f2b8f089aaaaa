// Fused BLAS + reduction kernels (role of reference lib/blas_quda.cu /
// lib/reduce_quda.cu functors, kernels/blas_core.cuh + reduce_core.cuh).
// Generic over the site layout via the accessor (Wilson 24-real sites and
// staggered 6-real sites share every functor); HALF (per-site norm)
// precision composes with every op; arithmetic always in Real (float for
// half), reductions accumulate double.
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

// finish mode: per_block=false -> one f64 atomic per block (fast path);
// per_block=true -> write partial[blockIdx] for a deterministic ordered
// host/torch sum (ref: QUDA_DETERMINISTIC_REDUCE, comm_quda.h:204)
__device__ __forceinline__ void block_atomic_add(double v, double *out,
                                                 bool per_block = false,
                                                 int slot_stride = 1,
                                                 int slot = 0) {
  __shared__ double partial[4];
  int lane = threadIdx.x & 63;
  int wave = threadIdx.x >> 6;
  v = wave_reduce(v);
  if (lane == 0) partial[wave] = v;
  __syncthreads();
  if (wave == 0) {
    double s = (lane < (int)(blockDim.x >> 6)) ? partial[lane] : 0.0;
    s = wave_reduce(s);
    if (lane == 0) {
      if (per_block)
        out[(long)blockIdx.x * slot_stride + slot] = s;
      else
        atomicAdd(out + slot, s);
    }
  }
}

#define GRID_STRIDE(g, n)                                                     \
  for (long g = (long)blockIdx.x * blockDim.x + threadIdx.x; g < (n);         \
       g += (long)gridDim.x * blockDim.x)

// y = a*x + y ; optional norm2(y) accumulation
template <typename A, bool NORM2>
__global__ __launch_bounds__(256) void k_axpy(
    typename A::R a, A x, A y, long sites, double *result, bool det) {
  using R = typename A::R;
  double acc = 0.0;
  GRID_STRIDE(g, sites) {
    cplx<R> xv[A::NCPLX], yv[A::NCPLX];
    x.load_v(xv, g);
    y.load_v(yv, g);
#pragma unroll
    for (int k = 0; k < A::NCPLX; ++k) {
      yv[k] = yv[k] + a * xv[k];
      if constexpr (NORM2)
        acc += (double)yv[k].re * yv[k].re + (double)yv[k].im * yv[k].im;
    }
    y.store_v(yv, g);
  }
  if constexpr (NORM2) block_atomic_add(acc, result, det, 2, 0);
}

// y = x + a*y
template <typename A>
__global__ __launch_bounds__(256) void k_xpay(
    A x, typename A::R a, A y, long sites) {
  using R = typename A::R;
  GRID_STRIDE(g, sites) {
    cplx<R> xv[A::NCPLX], yv[A::NCPLX];
    x.load_v(xv, g);
    y.load_v(yv, g);
#pragma unroll
    for (int k = 0; k < A::NCPLX; ++k) yv[k] = xv[k] + a * yv[k];
    y.store_v(yv, g);
  }
}

// y = a*x + b*y
template <typename A>
__global__ __launch_bounds__(256) void k_axpby(
    typename A::R a, A x, typename A::R b, A y, long sites) {
  using R = typename A::R;
  GRID_STRIDE(g, sites) {
    cplx<R> xv[A::NCPLX], yv[A::NCPLX];
    x.load_v(xv, g);
    y.load_v(yv, g);
#pragma unroll
    for (int k = 0; k < A::NCPLX; ++k) yv[k] = a * xv[k] + b * yv[k];
    y.store_v(yv, g);
  }
}

// y += (ar + i ai) * x
template <typename A>
__global__ __launch_bounds__(256) void k_caxpy(
    typename A::R ar, typename A::R ai, A x, A y, long sites) {
  using R = typename A::R;
  cplx<R> a{ar, ai};
  GRID_STRIDE(g, sites) {
    cplx<R> xv[A::NCPLX], yv[A::NCPLX];
    x.load_v(xv, g);
    y.load_v(yv, g);
#pragma unroll
    for (int k = 0; k < A::NCPLX; ++k) yv[k] = cfma(a, xv[k], yv[k]);
    y.store_v(yv, g);
  }
}

// y = (ar+i ai)*x + (br+i bi)*y
template <typename A>
__global__ __launch_bounds__(256) void k_caxpby(
    typename A::R ar, typename A::R ai, A x, typename A::R br,
    typename A::R bi, A y, long sites) {
  using R = typename A::R;
  cplx<R> a{ar, ai}, b{br, bi};
  GRID_STRIDE(g, sites) {
    cplx<R> xv[A::NCPLX], yv[A::NCPLX];
    x.load_v(xv, g);
    y.load_v(yv, g);
#pragma unroll
    for (int k = 0; k < A::NCPLX; ++k) yv[k] = cfma(a, xv[k], b * yv[k]);
    y.store_v(yv, g);
  }
}

// y = x - y ; returns norm2(y)
template <typename A>
__global__ __launch_bounds__(256) void k_xmy_norm2(
    A x, A y, long sites, double *result, bool det) {
  using R = typename A::R;
  double acc = 0.0;
  GRID_STRIDE(g, sites) {
    cplx<R> xv[A::NCPLX], yv[A::NCPLX];
    x.load_v(xv, g);
    y.load_v(yv, g);
#pragma unroll
    for (int k = 0; k < A::NCPLX; ++k) {
      yv[k] = xv[k] - yv[k];
      acc += (double)yv[k].re * yv[k].re + (double)yv[k].im * yv[k].im;
    }
    y.store_v(yv, g);
  }
  block_atomic_add(acc, result, det, 2, 0);
}

// x *= a
template <typename A>
__global__ __launch_bounds__(256) void k_scal(
    typename A::R a, A x, long sites) {
  using R = typename A::R;
  GRID_STRIDE(g, sites) {
    cplx<R> xv[A::NCPLX];
    x.load_v(xv, g);
#pragma unroll
    for (int k = 0; k < A::NCPLX; ++k) xv[k] = a * xv[k];
    x.store_v(xv, g);
  }
}

// reductions: norm2, re<x,y>, <x,y> (re+im)
template <typename A, int KIND>  // 0 norm2, 1 redot, 2 cdot
__global__ __launch_bounds__(256) void k_reduce(
    A x, A y, long sites, double *result, bool det) {
  using R = typename A::R;
  double acc = 0.0, acc2 = 0.0;
  GRID_STRIDE(g, sites) {
    cplx<R> xv[A::NCPLX], yv[A::NCPLX];
    x.load_v(xv, g);
    if constexpr (KIND != 0) y.load_v(yv, g);
#pragma unroll
    for (int k = 0; k < A::NCPLX; ++k) {
      if constexpr (KIND == 0) {
        acc += (double)xv[k].re * xv[k].re + (double)xv[k].im * xv[k].im;
      } else if constexpr (KIND == 1) {
        acc += (double)xv[k].re * yv[k].re + (double)xv[k].im * yv[k].im;
      } else {
        acc += (double)xv[k].re * yv[k].re + (double)xv[k].im * yv[k].im;
        acc2 += (double)xv[k].re * yv[k].im - (double)xv[k].im * yv[k].re;
      }
    }
  }
  block_atomic_add(acc, result, det, 2, 0);
  if constexpr (KIND == 2) block_atomic_add(acc2, result, det, 2, 1);
}

// precision conversion copy (ref: lib/copy_color_spinor_*.cu)
template <typename AD, typename AS>
__global__ __launch_bounds__(256) void k_convert(AD dst, AS src, long sites) {
  using RS = typename AS::R;
  using RD = typename AD::R;
  static_assert(AD::NCPLX == AS::NCPLX);
  GRID_STRIDE(g, sites) {
    cplx<RS> v[AS::NCPLX];
    src.load_v(v, g);
    cplx<RD> o[AD::NCPLX];
#pragma unroll
    for (int k = 0; k < AS::NCPLX; ++k) o[k] = {(RD)v[k].re, (RD)v[k].im};
    dst.store_v(o, g);
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
}  // namespace

// fused CG inner update (ref reduce_core.cuh tripleCGUpdate): one pass
// does the solution accumulation AND the residual update + its norm —
// two launches and one full residual re-read fewer per iteration
template <typename A>
__global__ __launch_bounds__(256) void k_triple_cg(
    typename A::R a, A p, A ap, A x, A r, long sites, double *result,
    bool det) {
  using R = typename A::R;
  double acc = 0.0;
  GRID_STRIDE(g, sites) {
    cplx<R> pv[A::NCPLX], av[A::NCPLX], xv[A::NCPLX], rv[A::NCPLX];
    p.load_v(pv, g);
    ap.load_v(av, g);
    x.load_v(xv, g);
    r.load_v(rv, g);
#pragma unroll
    for (int k = 0; k < A::NCPLX; ++k) {
      xv[k] = xv[k] + a * pv[k];
      rv[k] = rv[k] - a * av[k];
      acc += (double)rv[k].re * rv[k].re + (double)rv[k].im * rv[k].im;
    }
    x.store_v(xv, g);
    r.store_v(rv, g);
  }
  block_atomic_add(acc, result, det, 2, 0);
}

template <typename A>
static void blas_dispatch(const BlasCall &c, hipStream_t st) {
  using S = typename A::S;
  A x{(S *)c.x.data, (float *)c.x.norm, c.x.Vcb};
  A y{(S *)c.y.data, (float *)c.y.norm, c.y.Vcb};
  long n = c.sites;
  int gr = grid_for(n);
  using R = typename A::R;
  switch (c.op) {
    case BLAS_AXPY:
      hipLaunchKernelGGL((k_axpy<A, false>), dim3(gr), dim3(BLK), 0, st,
                         (R)c.a, x, y, n, nullptr, false);
      break;
    case BLAS_AXPY_NORM2:
      hipLaunchKernelGGL((k_axpy<A, true>), dim3(gr), dim3(BLK), 0, st,
                         (R)c.a, x, y, n, c.result, c.det);
      break;
    case BLAS_TRIPLE_CG: {
      A z{(S *)c.z.data, (float *)c.z.norm, c.z.Vcb};
      A w{(S *)c.w.data, (float *)c.w.norm, c.w.Vcb};
      hipLaunchKernelGGL((k_triple_cg<A>), dim3(gr), dim3(BLK), 0, st,
                         (R)c.a, x, y, z, w, n, c.result, c.det);
      break;
    }
    case BLAS_XPAY:
      hipLaunchKernelGGL((k_xpay<A>), dim3(gr), dim3(BLK), 0, st, x, (R)c.a, y, n);
      break;
    case BLAS_AXPBY:
      hipLaunchKernelGGL((k_axpby<A>), dim3(gr), dim3(BLK), 0, st, (R)c.a, x,
                         (R)c.b, y, n);
      break;
    case BLAS_CAXPY:
      hipLaunchKernelGGL((k_caxpy<A>), dim3(gr), dim3(BLK), 0, st, (R)c.a,
                         (R)c.b, x, y, n);
      break;
    case BLAS_CAXPBY:
      hipLaunchKernelGGL((k_caxpby<A>), dim3(gr), dim3(BLK), 0, st, (R)c.a,
                         (R)c.b, x, (R)c.c, (R)c.d, y, n);
      break;
    case BLAS_XMY_NORM2:
      hipLaunchKernelGGL((k_xmy_norm2<A>), dim3(gr), dim3(BLK), 0, st, x, y, n,
                         c.result, c.det);
      break;
    case BLAS_SCAL:
      hipLaunchKernelGGL((k_scal<A>), dim3(gr), dim3(BLK), 0, st, (R)c.a, x, n);
      break;
    case BLAS_NORM2:
      hipLaunchKernelGGL((k_reduce<A, 0>), dim3(gr), dim3(BLK), 0, st, x, y, n,
                         c.result, c.det);
      break;
    case BLAS_REDOT:
      hipLaunchKernelGGL((k_reduce<A, 1>), dim3(gr), dim3(BLK), 0, st, x, y, n,
                         c.result, c.det);
      break;
    case BLAS_CDOT:
      hipLaunchKernelGGL((k_reduce<A, 2>), dim3(gr), dim3(BLK), 0, st, x, y, n,
                         c.result, c.det);
      break;
  }
}

void launch_blas(const BlasCall &c, hipStream_t st) {
  if (c.ncomp == 6) {
    switch (c.prec) {
      case 0: blas_dispatch<StagAcc<PrecDouble>>(c, st); break;
      case 1: blas_dispatch<StagAcc<PrecSingle>>(c, st); break;
      case 2: blas_dispatch<StagAcc<PrecHalf>>(c, st); break;
      case 3: blas_dispatch<StagAcc<PrecQuarter>>(c, st); break;
    }
  } else {
    switch (c.prec) {
      case 0: blas_dispatch<SpinorAcc<PrecDouble>>(c, st); break;
      case 1: blas_dispatch<SpinorAcc<PrecSingle>>(c, st); break;
      case 2: blas_dispatch<SpinorAcc<PrecHalf>>(c, st); break;
      case 3: blas_dispatch<SpinorAcc<PrecQuarter>>(c, st); break;
    }
  }
}

template <template <typename> class AT, typename PD, typename PS>
static void conv(const BlasField &d, const BlasField &s, long sites, hipStream_t st) {
  auto da = AT<PD>{(typename PD::Store *)d.data, (float *)d.norm, d.Vcb};
  auto sa = AT<PS>{(typename PS::Store *)s.data, (float *)s.norm, s.Vcb};
  hipLaunchKernelGGL((k_convert<AT<PD>, AT<PS>>), dim3(grid_for(sites)),
                     dim3(BLK), 0, st, da, sa, sites);
}

template <template <typename> class AT>
static void conv_dispatch(const BlasField &dst, int pdst, const BlasField &src,
                          int psrc, long sites, hipStream_t st) {
  switch (pdst * 4 + psrc) {
    case 0 * 4 + 1: conv<AT, PrecDouble, PrecSingle>(dst, src, sites, st); break;
    case 0 * 4 + 2: conv<AT, PrecDouble, PrecHalf>(dst, src, sites, st); break;
    case 1 * 4 + 0: conv<AT, PrecSingle, PrecDouble>(dst, src, sites, st); break;
    case 1 * 4 + 2: conv<AT, PrecSingle, PrecHalf>(dst, src, sites, st); break;
    case 2 * 4 + 0: conv<AT, PrecHalf, PrecDouble>(dst, src, sites, st); break;
    case 2 * 4 + 1: conv<AT, PrecHalf, PrecSingle>(dst, src, sites, st); break;
    case 0 * 4 + 0: conv<AT, PrecDouble, PrecDouble>(dst, src, sites, st); break;
    case 1 * 4 + 1: conv<AT, PrecSingle, PrecSingle>(dst, src, sites, st); break;
    case 2 * 4 + 2: conv<AT, PrecHalf, PrecHalf>(dst, src, sites, st); break;
    case 0 * 4 + 3: conv<AT, PrecDouble, PrecQuarter>(dst, src, sites, st); break;
    case 1 * 4 + 3: conv<AT, PrecSingle, PrecQuarter>(dst, src, sites, st); break;
    case 2 * 4 + 3: conv<AT, PrecHalf, PrecQuarter>(dst, src, sites, st); break;
    case 3 * 4 + 0: conv<AT, PrecQuarter, PrecDouble>(dst, src, sites, st); break;
    case 3 * 4 + 1: conv<AT, PrecQuarter, PrecSingle>(dst, src, sites, st); break;
    case 3 * 4 + 2: conv<AT, PrecQuarter, PrecHalf>(dst, src, sites, st); break;
    case 3 * 4 + 3: conv<AT, PrecQuarter, PrecQuarter>(dst, src, sites, st); break;
  }
}

template <typename P>
using WilsonAccT = SpinorAcc<P, 24>;

void launch_convert(const BlasField &dst, int pdst, const BlasField &src,
                    int psrc, long sites, int ncomp, hipStream_t st) {
  if (ncomp == 6)
    conv_dispatch<StagAcc>(dst, pdst, src, psrc, sites, st);
  else
    conv_dispatch<WilsonAccT>(dst, pdst, src, psrc, sites, st);
}
