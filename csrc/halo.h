// Halo (ghost-zone) machinery for the Wilson-family stencils on MI355X.
// (role of reference include/kernels/dslash_pack.cuh + the ghost accessors in
//  color_spinor_field_order.h — redesigned: ghost faces hold spin-projected
//  half-spinors in a chunked SoA layout; comms ride RCCL/torch.distributed
//  from the Python layer, so this file is only pack kernels + accessors.)
//
// Ghost buffer layout per (dim mu, dir): [12/GW][Fcb][GW] Store reals where
// GW*sizeof(Store) is 16B (double/single) or 8B (half), Fcb = face cb volume.
// dir semantics ON THE RECEIVER: dir=1 ghost came from the +mu neighbor and
// feeds the forward hop; dir=0 came from -mu and feeds the backward hop.
// A SENDER packs send[mu][0] from its x_mu=0 face (consumed by the -mu
// neighbor as its dir=1 ghost) and send[mu][1] from its x_mu=L-1 face
// (consumed by +mu as dir=0). Projector for send[mu][dir] is
// proj_<mu>_<dir XOR dagger> — exactly what the consuming hop applies.
//
// Ghost face index (both ends agree; pure function of transverse coords):
//   flat3 = coords with dim mu deleted, lowest remaining dim fastest,
//   ghost_idx = flat3 >> 1   (bijective per parity since all extents even).
#pragma once

#include "common.h"
#include "generated/proj.h"

__device__ __forceinline__ long ghost_idx(const int x[4], int mu, const LatDims &d) {
  long f = 0;
#pragma unroll
  for (int i = 3; i >= 0; --i)
    if (i != mu) f = f * d.X[i] + x[i];
  return f >> 1;
}

// inverse: coords of the f-th parity-`parity` site on face x[mu] = fix
__device__ __forceinline__ void face_coords(int x[4], long f, int mu, int fix,
                                            const LatDims &d, int parity) {
  int rd0 = (mu == 0) ? 1 : 0;
  int rd1 = (mu <= 1) ? 2 : 1;
  int rd2 = (mu <= 2) ? 3 : 2;
  int h0 = d.X[rd0] >> 1;
  x[mu] = fix;
  int x0h = (int)(f % h0);
  long j = f / h0;
  x[rd1] = (int)(j % d.X[rd1]);
  x[rd2] = (int)(j / d.X[rd1]);
  int odd = (x[rd1] + x[rd2] + fix + parity + d.parity_offset) & 1;
  x[rd0] = 2 * x0h + odd;
}

template <typename Prec, int NCOMP = 12>
struct GhostAcc {
  using S = typename Prec::Store;
  using R = typename Prec::Real;
  static constexpr int GW =
      chunk_w<NCOMP, (Prec::W == 8) ? 4 : Prec::W>::value;
  static constexpr int NCH = NCOMP / GW;
  static constexpr int NCPLX = NCOMP / 2;
  const S *buf[8];      // [2*mu+dir]; null when dim not partitioned
  const float *nrm[8];  // half only
  long Fcb[4];
  int mask;             // bit mu set => dim mu partitioned
  int depth = 1;        // ghost layers per (mu,dir) (3 for Naik long links)

  __device__ __forceinline__ bool active(int mu) const { return (mask >> mu) & 1; }

  __device__ __forceinline__ void load_v(cplx<R> out[NCPLX], int mu, int dir,
                                         long f) const {
    const S *b = buf[2 * mu + dir];
    S tmp[NCOMP];
    long stride = (long)depth * Fcb[mu];
#pragma unroll
    for (int ch = 0; ch < NCH; ++ch)
      load_chunk<S, GW>(b + ((long)ch * stride + f) * GW, tmp + ch * GW);
    R scale = (R)1;
    if constexpr (Prec::has_norm) scale = nrm[2 * mu + dir][f];
#pragma unroll
    for (int k = 0; k < NCPLX; ++k)
      {
        R re, im;
        qa_tor_pair<R, S>(tmp + 2 * k, re, im);
        out[k] = {scale * re, scale * im};
      }
  }

  // Wilson half-spinor view (NCOMP == 12)
  __device__ __forceinline__ void load(cplx<R> (&h)[2][3], int mu, int dir,
                                       long f) const {
    static_assert(NCOMP == 12);
    load_v(reinterpret_cast<cplx<R> *>(h), mu, dir, f);
  }

  // batched-halo view: rhs slice r of a [n_rhs][NCH][depth*Fcb][GW]
  // buffer (parallel/halo.py BatchSpinorHalo — buf[] points at slice 0)
  __device__ __forceinline__ void load_r(cplx<R> (&h)[2][3], int mu, int dir,
                                         long f, int r) const {
    static_assert(NCOMP == 12);
    const S *b = buf[2 * mu + dir];
    long stride = (long)depth * Fcb[mu];
    b += (long)r * NCH * stride * GW;
    S tmp[NCOMP];
#pragma unroll
    for (int ch = 0; ch < NCH; ++ch)
      load_chunk<S, GW>(b + ((long)ch * stride + f) * GW, tmp + ch * GW);
    R scale = (R)1;
    if constexpr (Prec::has_norm) scale = nrm[2 * mu + dir][(long)r * stride + f];
    auto *out = reinterpret_cast<cplx<R> *>(h);
#pragma unroll
    for (int k = 0; k < NCPLX; ++k)
      {
        R re, im;
        qa_tor_pair<R, S>(tmp + 2 * k, re, im);
        out[k] = {scale * re, scale * im};
      }
  }
};

// write NCOMP/2 complex values into a send buffer slot
template <typename Prec, int NCOMP>
__device__ __forceinline__ void ghost_store_v(
    typename Prec::Store *buf, float *nrm, long Fcb, long f,
    const cplx<typename Prec::Real> *h) {
  using S = typename Prec::Store;
  using R = typename Prec::Real;
  constexpr int GW = GhostAcc<Prec, NCOMP>::GW;
  constexpr int NCH = GhostAcc<Prec, NCOMP>::NCH;
  constexpr int NCPLX = NCOMP / 2;
  S tmp[NCOMP];
  if constexpr (Prec::has_norm) {
    R m = (R)0;
#pragma unroll
    for (int k = 0; k < NCPLX; ++k)
      m = fmax(m, fmax(fabs(h[k].re), fabs(h[k].im)));
    nrm[f] = m;
    R inv = m > (R)0 ? (R)1 / m : (R)0;
#pragma unroll
    for (int k = 0; k < NCPLX; ++k) {
      tmp[2 * k] = qa_tos<S>(h[k].re * inv);
      tmp[2 * k + 1] = qa_tos<S>(h[k].im * inv);
    }
  } else {
#pragma unroll
    for (int k = 0; k < NCPLX; ++k) {
      tmp[2 * k] = (S)h[k].re;
      tmp[2 * k + 1] = (S)h[k].im;
    }
  }
#pragma unroll
  for (int ch = 0; ch < NCH; ++ch)
    store_chunk<S, GW>(buf + ((long)ch * Fcb + f) * GW, tmp + ch * GW);
}

template <typename Prec>
__device__ __forceinline__ void ghost_store(typename Prec::Store *buf, float *nrm,
                                            long Fcb, long f,
                                            const cplx<typename Prec::Real> h[2][3]) {
  ghost_store_v<Prec, 12>(buf, nrm, Fcb, f,
                          reinterpret_cast<const cplx<typename Prec::Real> *>(h));
}

// Pack one face of `in` (the dslash input spinor, at parity `parity`) into a
// send buffer: h = proj_<MU>_<S01>(psi) over face x_MU = (EDGE ? X-1 : 0).
// One thread per face site.
template <typename Prec, int MU, int S01, bool EDGE>
__global__ __launch_bounds__(256) void k_pack_face(
    typename Prec::Store *dst, float *dst_nrm, SpinorAcc<Prec> in, LatDims d,
    int parity, long Fcb) {
  using R = typename Prec::Real;
  long f = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (f >= Fcb) return;
  int xc[4];
  face_coords(xc, f, MU, EDGE ? d.X[MU] - 1 : 0, d, parity);
  long i = cb_from_coords(xc, d);
  cplx<R> p[4][3], h[2][3];
  in.load(p, i);
  if constexpr (MU == 0) { if constexpr (S01 == 0) proj_0_0(h, p); else proj_0_1(h, p); }
  if constexpr (MU == 1) { if constexpr (S01 == 0) proj_1_0(h, p); else proj_1_1(h, p); }
  if constexpr (MU == 2) { if constexpr (S01 == 0) proj_2_0(h, p); else proj_2_1(h, p); }
  if constexpr (MU == 3) { if constexpr (S01 == 0) proj_3_0(h, p); else proj_3_1(h, p); }
  ghost_store<Prec>(dst, dst_nrm, Fcb, f, h);
}
