// Wilson / Wilson-clover dslash stencil for MI355X (gfx950).
// (role of reference include/kernels/dslash_wilson.cuh applyWilson +
//  dslash_wilson_clover_preconditioned.cuh — redesigned: one thread per
//  output site, grid-stride, 16B vector loads, fused clover/xpay epilogues)
//
// Kernel modes:
//   PLAIN      : out = [x +] a * (D in)            (XPAY: x term present)
//   CLOV_POST  : out = [x +] a * (A_clov (D in))   (A = packed site matrix,
//                usually the inverse: even-odd preconditioned operator)
//   CLOV_X     : out = A_clov x + a * (D in)       (full clover M, a = -kappa)
//
// so DiracWilsonPC::M = 2 launches, DiracCloverPC::M = 2 launches.
#pragma once

#include "common.h"
#include "generated/proj.h"
#include "halo.h"

// epilogue modes (b = (b_re, b_im) twist scalar, T(b) v = b_re v + i b_im g5 v):
//   PLAIN      : out = [x +] a * (D in)
//   CLOV_POST  : out = [x +] a * (A (D in))
//   CLOV_X     : out = A x + a * (D in)
//   TWIST_POST : out = [x +] a * (T(b) (D in))   (twisted-mass PC: T = A^-1)
//   TWIST_X    : out = T(b) x + a * (D in)       (twisted-mass full op)
//   CLOVTW_X   : out = (A + i b_im g5) x + a * (D in)  (twisted-clover full)
enum CloverMode { PLAIN = 0, CLOV_POST = 1, CLOV_X = 2, TWIST_POST = 3,
                  TWIST_X = 4, CLOVTW_X = 5 };

// v <- b_re v + i b_im g5 v (DeGrand-Rossi g5 = diag(1,1,-1,-1))
template <typename R>
__device__ __forceinline__ void twist_mul(cplx<R> v[4][3], R br, R bi) {
#pragma unroll
  for (int s = 0; s < 4; ++s) {
    R bs = (s < 2) ? bi : -bi;
#pragma unroll
    for (int c = 0; c < 3; ++c) {
      cplx<R> t = v[s][c];
      v[s][c] = {br * t.re - bs * t.im, br * t.im + bs * t.re};
    }
  }
}


// packed clover (fields/clover.py): per site 72 reals =
// 2 chirality blocks x (6 diag + 15 lower-tri complex)
template <typename Prec>
struct CloverAcc {
  using S = typename Prec::Store;
  using R = typename Prec::Real;
  // widest chunk dividing the 72-real site (quarter: 16 does not divide
  // 72 -> 8-byte chunks, mirroring fields/layout.py chunk_width)
  static constexpr int W = chunk_w<72, Prec::W>::value;
  static constexpr int NCH = 72 / W;
  const S *data;  // [parity][NCH][V][W]
  long V;

  __device__ __forceinline__ void load(R diag[2][6], cplx<R> tri[2][15],
                                       int parity, long i) const {
    S tmp[72];
    const S *base = data + ((long)parity * NCH * V + i) * W;
#pragma unroll
    for (int ch = 0; ch < NCH; ++ch)
      load_chunk<S, W>(base + (long)ch * V * W, tmp + ch * W);
#pragma unroll
    for (int b = 0; b < 2; ++b) {
#pragma unroll
      for (int k = 0; k < 6; ++k) diag[b][k] = qa_tor<R>(tmp[36 * b + k]);
#pragma unroll
      for (int k = 0; k < 15; ++k) {
        R re, im;
        qa_tor_pair<R, S>(tmp + 36 * b + 6 + 2 * k, re, im);
        tri[b][k] = {re, im};
      }
    }
  }
};

// apply packed hermitian 2x(6x6) clover to a spinor held as [4][3]
// chirality block b covers spins {2b, 2b+1}; within-block index = s*3+c (s=0,1)
template <typename R>
__device__ __forceinline__ void clover_mul(cplx<R> out[4][3], const R diag[2][6],
                                           const cplx<R> tri[2][15],
                                           const cplx<R> in[4][3]) {
  // lower-tri packing: k runs over (i,j), i>j, row-major: (1,0),(2,0),(2,1),...
#pragma unroll
  for (int b = 0; b < 2; ++b) {
    cplx<R> v[6];
#pragma unroll
    for (int k = 0; k < 6; ++k) v[k] = in[2 * b + k / 3][k % 3];
    cplx<R> r[6];
#pragma unroll
    for (int k = 0; k < 6; ++k) r[k] = diag[b][k] * v[k];
    int k = 0;
#pragma unroll
    for (int i = 1; i < 6; ++i)
#pragma unroll
      for (int j = 0; j < i; ++j, ++k) {
        r[i] = cfma(tri[b][k], v[j], r[i]);          // A[i][j] v[j]
        r[j] = cfma_conj(tri[b][k], v[i], r[j]);     // conj(A[i][j]) v[i]
      }
#pragma unroll
    for (int kk = 0; kk < 6; ++kk) out[2 * b + kk / 3][kk % 3] = r[kk];
  }
}

// LBW = 0: 256-thread cap, allocator unconstrained (measured 222 VGPRs ->
// 2 waves/SIMD). LBW > 0: 64-thread workgroups with a minimum of LBW waves
// per SIMD — caps the register allocation (512/LBW) to raise occupancy on
// this latency-bound kernel (profiles/r01_dslash_pmc_analysis.md).
template <typename Prec, int RECON, bool DAG, int MODE, bool XPAY, int KT = KT_LOCAL,
          int LBW = 0>
__global__ __launch_bounds__(LBW ? 64 : 256, LBW ? LBW : 1) void k_dslash_wilson(
    SpinorAcc<Prec> out, SpinorAcc<Prec> in, GaugeAcc<Prec, RECON> g,
    CloverAcc<Prec> clov, LatDims d, int parity, typename Prec::Real a,
    SpinorAcc<Prec> x, GhostAcc<Prec> gh, typename Prec::Real br,
    typename Prec::Real bi) {
  using R = typename Prec::Real;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= d.Vcb) return;
  int xc[4];
  coords_from_cb(xc, i, d, parity);

  cplx<R> acc[4][3];
#pragma unroll
  for (int s = 0; s < 4; ++s)
#pragma unroll
    for (int c = 0; c < 3; ++c) acc[s][c] = {(R)0, (R)0};

  cplx<R> p[4][3], h[8][2][3], uh[2][3], U[3][3];
  // proj/recon tables encode 2P (generate_proj.py); the stencil needs P
  const R one = (R)0.5;

  bool bnd = false;  // interior: site has >=1 hop deferred to EXTERIOR
  if constexpr (KT == KT_INTERIOR) {
#pragma unroll
    for (int m = 0; m < 4; ++m)
      if (gh.active(m) && (xc[m] == 0 || xc[m] == d.X[m] - 1)) bnd = true;
  }
  bool skip[8];
#pragma unroll
  for (int k = 0; k < 8; ++k) skip[k] = false;
  // -mu neighbor index, kept for phase 2: the backward link U_mu(x-mu) is
  // read from the NEIGHBOR's forward slot instead of this site's bwd slot.
  // The stencil layout stores every link twice (fwd slot of x == bwd slot
  // of x+mu); reading only fwd slots halves the dslash's UNIQUE gauge
  // traffic — neighbor fwd-slot lines are shared through L2 exactly like
  // the neighbor spinor loads. Boundary-crossing hops (jm < 0) still read
  // the bwd slot, which holds the -mu RANK's link from the load-time
  // exchange (fields/gauge.py from_complex).
  long jm[4];
#pragma unroll
  for (int k = 0; k < 4; ++k) jm[k] = -1;

  // Phase 1: gather + spin-project all 8 neighbor half-spinors first — the
  // loads are independent, so the wave has 8 spinor fetches in flight
  // instead of a serialized load->project->multiply chain per direction
  // (the dslash is latency-bound at ~30% of HBM peak otherwise; measured
  // FETCH_SIZE is already near-minimal).
#define QA_GATHER(MU)                                                     \
  {                                                                       \
    bool cross_p = KT != KT_LOCAL && gh.active(MU) && xc[MU] == d.X[MU] - 1; \
    if (KT == KT_INTERIOR && cross_p) {                                   \
      skip[2 * MU] = true;                                                \
    } else if (KT == KT_FUSED && cross_p) {                               \
      gh.load(h[2 * MU], MU, 1, ghost_idx(xc, MU, d));                    \
    } else {                                                              \
      long j = neighbor_cb(xc, MU, +1, d);                                \
      in.load(p, j);                                                      \
      if constexpr (!DAG) proj_##MU##_0(h[2 * MU], p);                    \
      else proj_##MU##_1(h[2 * MU], p);                                   \
    }                                                                     \
    bool cross_m = KT != KT_LOCAL && gh.active(MU) && xc[MU] == 0;        \
    if (KT == KT_INTERIOR && cross_m) {                                   \
      skip[2 * MU + 1] = true;                                            \
    } else if (KT == KT_FUSED && cross_m) {                               \
      gh.load(h[2 * MU + 1], MU, 0, ghost_idx(xc, MU, d));                \
    } else {                                                              \
      long j = neighbor_cb(xc, MU, -1, d);                                \
      jm[MU] = j;                                                         \
      in.load(p, j);                                                      \
      if constexpr (!DAG) proj_##MU##_1(h[2 * MU + 1], p);                \
      else proj_##MU##_0(h[2 * MU + 1], p);                               \
    }                                                                     \
  }

  // Phase 2 macro: gauge multiplies + spin reconstruction (fwd link at own
  // site; bwd link from the -mu neighbor's fwd slot, see jm above)
#define QA_MUL(MU)                                                        \
  {                                                                       \
    if (!skip[2 * MU]) {                                                  \
      g.template load<MU>(U, i);                                          \
      su3_mul_half(uh, U, h[2 * MU]);                                     \
      if constexpr (!DAG) recon_##MU##_0(acc, uh, one);                   \
      else recon_##MU##_1(acc, uh, one);                                  \
    }                                                                     \
    if (!skip[2 * MU + 1]) {                                              \
      if (jm[MU] >= 0) g.template load_o<MU>(U, jm[MU]);                  \
      else g.template load<4 + MU>(U, i);                                 \
      su3_dagmul_half(uh, U, h[2 * MU + 1]);                              \
      if constexpr (!DAG) recon_##MU##_1(acc, uh, one);                   \
      else recon_##MU##_0(acc, uh, one);                                  \
    }                                                                     \
  }

  if constexpr (LBW == 0) {
    // all 8 gathers in flight, then all multiplies (max load ILP; ~210
    // VGPRs -> 2 waves/SIMD)
    QA_GATHER(0)
    QA_GATHER(1)
    QA_GATHER(2)
    QA_GATHER(3)
    QA_MUL(0)
    QA_MUL(1)
    QA_MUL(2)
    QA_MUL(3)
  } else {
    // split-gather: two 4-neighbor rounds halve the live half-spinor
    // state so the LBW-wave occupancy cap is met without spilling —
    // trades per-wave ILP for thread-level parallelism.
    QA_GATHER(0)
    QA_GATHER(1)
    QA_MUL(0)
    QA_MUL(1)
    // pin the round split: without this the scheduler hoists round-2 loads
    // into round 1 and the live state spills right back to the 8-gather shape
    __builtin_amdgcn_sched_barrier(0);
    QA_GATHER(2)
    QA_GATHER(3)
    QA_MUL(2)
    QA_MUL(3)
  }
#undef QA_GATHER
#undef QA_MUL

  // boundary sites under CLOV_POST defer the whole epilogue: store raw sum
  if constexpr (KT == KT_INTERIOR && MODE == CLOV_POST) {
    if (bnd) {
      out.store(acc, i);
      return;
    }
  }

  if constexpr (MODE == CLOV_POST) {
    R diag[2][6];
    cplx<R> tri[2][15];
    clov.load(diag, tri, parity, i);
    cplx<R> tmp[4][3];
    clover_mul(tmp, diag, tri, acc);
#pragma unroll
    for (int s = 0; s < 4; ++s)
#pragma unroll
      for (int c = 0; c < 3; ++c) acc[s][c] = tmp[s][c];
  }
  if constexpr (MODE == TWIST_POST) twist_mul(acc, br, bi);

  if constexpr (XPAY || MODE == CLOV_X || MODE == TWIST_X || MODE == CLOVTW_X) {
    cplx<R> xv[4][3];
    x.load(xv, i);
    if constexpr (MODE == CLOV_X || MODE == CLOVTW_X) {
      R diag[2][6];
      cplx<R> tri[2][15];
      clov.load(diag, tri, parity, i);
      cplx<R> Ax[4][3];
      clover_mul(Ax, diag, tri, xv);
      if constexpr (MODE == CLOVTW_X) {
        twist_mul(xv, (R)0, bi);  // i b_im g5 x
#pragma unroll
        for (int s = 0; s < 4; ++s)
#pragma unroll
          for (int c = 0; c < 3; ++c) Ax[s][c] += xv[s][c];
      }
#pragma unroll
      for (int s = 0; s < 4; ++s)
#pragma unroll
        for (int c = 0; c < 3; ++c) acc[s][c] = Ax[s][c] + a * acc[s][c];
    } else if constexpr (MODE == TWIST_X) {
      twist_mul(xv, br, bi);
#pragma unroll
      for (int s = 0; s < 4; ++s)
#pragma unroll
        for (int c = 0; c < 3; ++c) acc[s][c] = xv[s][c] + a * acc[s][c];
    } else {
#pragma unroll
      for (int s = 0; s < 4; ++s)
#pragma unroll
        for (int c = 0; c < 3; ++c) acc[s][c] = xv[s][c] + a * acc[s][c];
    }
  } else {
    if (a != (R)1) {
#pragma unroll
      for (int s = 0; s < 4; ++s)
#pragma unroll
        for (int c = 0; c < 3; ++c) acc[s][c] = a * acc[s][c];
    }
  }

  out.store(acc, i);
}

// EXTERIOR: one owner thread per boundary site adds all ghost-hop
// contributions and completes deferred epilogues (see DslashKT above).
// Thread space: faces in (mu asc, edge 0,1) order over active dims.
template <typename Prec, int RECON, bool DAG, int MODE, bool XPAY>
__global__ __launch_bounds__(256) void k_dslash_wilson_exterior(
    SpinorAcc<Prec> out, SpinorAcc<Prec> in, GaugeAcc<Prec, RECON> g,
    CloverAcc<Prec> clov, LatDims d, int parity, typename Prec::Real a,
    SpinorAcc<Prec> x, GhostAcc<Prec> gh, long n_threads,
    typename Prec::Real br, typename Prec::Real bi) {
  using R = typename Prec::Real;
  long tid = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (tid >= n_threads) return;
  int mu = -1, edge = 0;
  long f = tid;
#pragma unroll
  for (int m = 0; m < 4; ++m) {
    if (mu < 0 && gh.active(m)) {
      if (f < 2 * gh.Fcb[m]) {
        mu = m;
        edge = f >= gh.Fcb[m];
        if (edge) f -= gh.Fcb[m];
      } else {
        f -= 2 * gh.Fcb[m];
      }
    }
  }
  if (mu < 0) return;
  int xc[4];
  face_coords(xc, f, mu, edge ? d.X[mu] - 1 : 0, d, parity);
  // ownership: the first (m, edge) face containing this site handles ALL
  // of its ghost hops (avoids read-modify-write races on corner sites)
  int key = 2 * mu + edge;
#pragma unroll
  for (int m = 0; m < 4; ++m) {
    if (gh.active(m)) {
      if (xc[m] == 0 && 2 * m < key) return;
      if (xc[m] == d.X[m] - 1 && 2 * m + 1 < key) return;
    }
  }
  long i = cb_from_coords(xc, d);

  cplx<R> acc[4][3];
#pragma unroll
  for (int s = 0; s < 4; ++s)
#pragma unroll
    for (int c = 0; c < 3; ++c) acc[s][c] = {(R)0, (R)0};
  cplx<R> h[2][3], uh[2][3], U[3][3];
  const R one = (R)0.5;

#define QA_EXT_DIR(MU)                                                    \
  if (gh.active(MU)) {                                                    \
    if (xc[MU] == d.X[MU] - 1) {                                          \
      gh.load(h, MU, 1, ghost_idx(xc, MU, d));                            \
      g.template load<MU>(U, i);                                          \
      su3_mul_half(uh, U, h);                                             \
      if constexpr (!DAG) recon_##MU##_0(acc, uh, one);                   \
      else recon_##MU##_1(acc, uh, one);                                  \
    }                                                                     \
    if (xc[MU] == 0) {                                                    \
      gh.load(h, MU, 0, ghost_idx(xc, MU, d));                            \
      g.template load<4 + MU>(U, i);                                      \
      su3_dagmul_half(uh, U, h);                                          \
      if constexpr (!DAG) recon_##MU##_1(acc, uh, one);                   \
      else recon_##MU##_0(acc, uh, one);                                  \
    }                                                                     \
  }

  QA_EXT_DIR(0)
  QA_EXT_DIR(1)
  QA_EXT_DIR(2)
  QA_EXT_DIR(3)
#undef QA_EXT_DIR

  cplx<R> prev[4][3];
  out.load(prev, i);
  if constexpr (MODE == CLOV_POST) {
    // prev = raw interior hop sum; complete sum, then epilogue
#pragma unroll
    for (int s = 0; s < 4; ++s)
#pragma unroll
      for (int c = 0; c < 3; ++c) acc[s][c] += prev[s][c];
    R diag[2][6];
    cplx<R> tri[2][15];
    clov.load(diag, tri, parity, i);
    cplx<R> tmp[4][3];
    clover_mul(tmp, diag, tri, acc);
    if constexpr (XPAY) {
      cplx<R> xv[4][3];
      x.load(xv, i);
#pragma unroll
      for (int s = 0; s < 4; ++s)
#pragma unroll
        for (int c = 0; c < 3; ++c) acc[s][c] = xv[s][c] + a * tmp[s][c];
    } else {
#pragma unroll
      for (int s = 0; s < 4; ++s)
#pragma unroll
        for (int c = 0; c < 3; ++c) acc[s][c] = a * tmp[s][c];
    }
  } else {
    // prev already carries the (affine) epilogue on the partial sum;
    // T(b) is linear and site-diagonal, so TWIST_POST twists the ghost
    // contribution alone
    if constexpr (MODE == TWIST_POST) twist_mul(acc, br, bi);
#pragma unroll
    for (int s = 0; s < 4; ++s)
#pragma unroll
      for (int c = 0; c < 3; ++c) acc[s][c] = prev[s][c] + a * acc[s][c];
  }
  out.store(acc, i);
}

// standalone clover apply: out = A(parity) in
template <typename Prec>
__global__ __launch_bounds__(256) void k_clover_apply(
    SpinorAcc<Prec> out, SpinorAcc<Prec> in, CloverAcc<Prec> clov,
    int parity_slot, long Vcb) {
  using R = typename Prec::Real;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= Vcb) return;
  cplx<R> v[4][3], r[4][3];
  R diag[2][6];
  cplx<R> tri[2][15];
  in.load(v, i);
  clov.load(diag, tri, parity_slot, i);
  clover_mul(r, diag, tri, v);
  out.store(r, i);
}

// ---------------------------------------------------------------------------
// LDS-tiled Wilson dslash (half/quarter, local-only).
//
// The gather-first kernel above is LOAD-ISSUE bound on MI355X (~13 B/cyc/
// CU at the mixed-load ceiling, profiles/r02_dslash_final.md): every
// neighbor spinor body is fetched from global memory TWICE per
// checkerboard apply (once by each adjacent output site), and all fetches
// compete for the same vector-load issue port. This variant stages the
// whole in-spinor halo tile through LDS instead: one workgroup owns a
// BX x BY x BZ x BT lexicographic tile (256 output sites of one parity),
// cooperatively loads the (B+2)^-extended in-parity halo (1080 sites,
// RAW stored chunks + per-site norm = 56 KB at half -> 2 workgroups/CU),
// then serves all 8 neighbor reads per site from the LDS pipe, which
// issues independently of the global-load port. Unique global spinor
// traffic drops ~1.9x and the spinor share of the vector-load issue
// stream moves off the critical port entirely; gauge links still stream
// from global (fwd-slot reads as above).
// (role of reference kernels/dslash_wilson.cuh; the reference deliberately
// has no shared-memory dslash — on NVIDIA the texture path suffices. On
// CDNA4 the LDS pipe is the only way past the load-issue ceiling.)
// ---------------------------------------------------------------------------
struct LdsTile {
  // 4^4 tile: 128 output sites/WG, ~34 KB LDS at half -> 4 workgroups/CU.
  // The compute phase's gauge loads serialize within one wave (the
  // allocation sits exactly at the 2-wave 256-VGPR line, so the
  // scheduler cannot hoist loads across directions); four independent
  // workgroups per CU restore the memory-level parallelism instead.
  static constexpr int BX = 4, BY = 4, BZ = 4, BT = 8;
  static constexpr int EX = BX + 2, EY = BY + 2, EZ = BZ + 2, ET = BT + 2;
  static constexpr int NROW = EY * EZ * ET;              // rows of EX sites
  static constexpr int NSLOT = NROW * (EX / 2);          // in-parity sites
  static constexpr int NOUT = BX * BY * BZ * BT / 2;     // threads/WG
};

template <typename Prec, int RECON, bool DAG, int MODE, bool XPAY>
__global__ __launch_bounds__(LdsTile::NOUT, 1) void k_dslash_wilson_lds(
    SpinorAcc<Prec> out, SpinorAcc<Prec> in, GaugeAcc<Prec, RECON> g,
    CloverAcc<Prec> clov, LatDims d, int parity, typename Prec::Real a,
    SpinorAcc<Prec> x, typename Prec::Real br, typename Prec::Real bi) {
  using R = typename Prec::Real;
  using S = typename Prec::Store;
  static_assert(Prec::has_norm, "LDS dslash: half/quarter only (raw-chunk staging)");
  constexpr int W = SpinorAcc<Prec>::W;     // 8 for half & quarter
  constexpr int NCH = SpinorAcc<Prec>::NCH; // 3
  using T = LdsTile;

  // 16-byte alignment is REQUIRED: chunks move via ds_read/write_b128
  __shared__ alignas(16) S lds[NCH][T::NSLOT][W];
  __shared__ float lnrm[T::NSLOT];
  int qa_xc[4];  // this thread's output-site coords

  // tile origin from block id (x-fastest tile raster)
  int ntx = d.X[0] / T::BX, nty = d.X[1] / T::BY, ntz = d.X[2] / T::BZ;
  long bid = blockIdx.x;
  int ox = (int)(bid % ntx) * T::BX; bid /= ntx;
  int oy = (int)(bid % nty) * T::BY; bid /= nty;
  int oz = (int)(bid % ntz) * T::BZ;
  int ot = (int)(bid / ntz) * T::BT;
  const int ipar = 1 - parity;
  const int S0 = ox + oy + oz + ot + d.parity_offset;

  // ---- output site of this thread (needed first: the raw gauge
  //      preloads below overlap the cooperative LDS fill) ----
  {
    int k0 = threadIdx.x % (T::BX / 2), r1 = threadIdx.x / (T::BX / 2);
    int ly = r1 % T::BY; int rr1 = r1 / T::BY;
    int lz = rr1 % T::BZ, lt = rr1 / T::BZ;
    int rp = (parity + S0 + ly + lz + lt) & 1;
    qa_xc[0] = ox + rp + 2 * k0; qa_xc[1] = oy + ly;
    qa_xc[2] = oz + lz; qa_xc[3] = ot + lt;
  }
  const long i = cb_from_coords(qa_xc, d);

  // ---- cooperative halo load: slot s covers ext row r = s / (EX/2),
  //      within-row k = s % (EX/2); in-parity x positions are r0 + 2k ----
#pragma unroll 1
  for (int s = threadIdx.x; s < T::NSLOT; s += blockDim.x) {
    int k = s % (T::EX / 2), r = s / (T::EX / 2);
    int ey = r % T::EY, rr = r / T::EY;
    int ez = rr % T::EZ, et = rr / T::EZ;
    int r0 = (ipar + S0 + ey + ez + et) & 1;
    int gx = ox - 1 + r0 + 2 * k, gy = oy - 1 + ey, gz = oz - 1 + ez,
        gt = ot - 1 + et;
    if (gx < 0) gx += d.X[0]; else if (gx >= d.X[0]) gx -= d.X[0];
    if (gy < 0) gy += d.X[1]; else if (gy >= d.X[1]) gy -= d.X[1];
    if (gz < 0) gz += d.X[2]; else if (gz >= d.X[2]) gz -= d.X[2];
    if (gt < 0) gt += d.X[3]; else if (gt >= d.X[3]) gt -= d.X[3];
    int xc2[4] = {gx, gy, gz, gt};
    long j = cb_from_coords(xc2, d);
    long base = j * W;  // parity-sliced accessor: chunk_base(j<V) = j*W
#pragma unroll
    for (int ch = 0; ch < NCH; ++ch)
      load_chunk<S, W>(in.data + base + (long)ch * in.V * W, lds[ch][s]);
    lnrm[s] = in.norm[j];
  }
  __syncthreads();

  cplx<R> acc[4][3];
#pragma unroll
  for (int s = 0; s < 4; ++s)
#pragma unroll
    for (int c = 0; c < 3; ++c) acc[s][c] = {(R)0, (R)0};
  cplx<R> p[4][3], h[2][3], uh[2][3], U[3][3];
  const R one = (R)0.5;

  // LDS slot of the neighbor at ext coords; the coords always hold an
  // in-parity site (macro, not lambda: capture state stays dead)
#define QA_LDS_READ(EXC, EYC, EZC, ETC)                                   \
  {                                                                       \
    int slot_ = (((ETC) * T::EZ + (EZC)) * T::EY + (EYC)) * (T::EX / 2) + \
                ((EXC) >> 1);                                             \
    S tmp_[24];                                                           \
    _Pragma("unroll")                                                     \
    for (int ch_ = 0; ch_ < NCH; ++ch_)                                   \
      load_chunk<S, W>(lds[ch_][slot_], tmp_ + ch_ * W);                  \
    R sc_ = lnrm[slot_];                                                  \
    _Pragma("unroll")                                                     \
    for (int kk_ = 0; kk_ < 12; ++kk_) {                                  \
      R re_, im_;                                                         \
      qa_tor_pair<R, S>(tmp_ + 2 * kk_, re_, im_);                        \
      p[kk_ / 3][kk_ % 3] = {sc_ * re_, sc_ * im_};                       \
    }                                                                     \
  }

  const int e0 = qa_xc[0] - ox + 1, e1 = qa_xc[1] - oy + 1,
            e2 = qa_xc[2] - oz + 1, e3 = qa_xc[3] - ot + 1;

#define QA_LDS_DIR(MU, EXP, EXM)                                          \
  {                                                                       \
    EXP;                                                                  \
    if constexpr (!DAG) proj_##MU##_0(h, p);                              \
    else proj_##MU##_1(h, p);                                             \
    g.template load<MU>(U, i);                                            \
    su3_mul_half(uh, U, h);                                               \
    if constexpr (!DAG) recon_##MU##_0(acc, uh, one);                     \
    else recon_##MU##_1(acc, uh, one);                                    \
    EXM;                                                                  \
    if constexpr (!DAG) proj_##MU##_1(h, p);                              \
    else proj_##MU##_0(h, p);                                             \
    g.template load_o<MU>(U, neighbor_cb(qa_xc, MU, -1, d));             \
    su3_dagmul_half(uh, U, h);                                            \
    if constexpr (!DAG) recon_##MU##_1(acc, uh, one);                     \
    else recon_##MU##_0(acc, uh, one);                                    \
  }

  QA_LDS_DIR(0, QA_LDS_READ(e0 + 1, e1, e2, e3), QA_LDS_READ(e0 - 1, e1, e2, e3))
  QA_LDS_DIR(1, QA_LDS_READ(e0, e1 + 1, e2, e3), QA_LDS_READ(e0, e1 - 1, e2, e3))
  QA_LDS_DIR(2, QA_LDS_READ(e0, e1, e2 + 1, e3), QA_LDS_READ(e0, e1, e2 - 1, e3))
  QA_LDS_DIR(3, QA_LDS_READ(e0, e1, e2, e3 + 1), QA_LDS_READ(e0, e1, e2, e3 - 1))
#undef QA_LDS_DIR
#undef QA_LDS_READ

  if constexpr (MODE == CLOV_POST) {
    R diag[2][6];
    cplx<R> tri[2][15];
    clov.load(diag, tri, parity, i);
    cplx<R> tmp[4][3];
    clover_mul(tmp, diag, tri, acc);
#pragma unroll
    for (int s = 0; s < 4; ++s)
#pragma unroll
      for (int c = 0; c < 3; ++c) acc[s][c] = tmp[s][c];
  }
  if constexpr (MODE == TWIST_POST) twist_mul(acc, br, bi);
  if constexpr (XPAY) {
    cplx<R> xv[4][3];
    x.load(xv, i);
#pragma unroll
    for (int s = 0; s < 4; ++s)
#pragma unroll
      for (int c = 0; c < 3; ++c) acc[s][c] = xv[s][c] + a * acc[s][c];
  } else {
    if (a != (R)1) {
#pragma unroll
      for (int s = 0; s < 4; ++s)
#pragma unroll
        for (int c = 0; c < 3; ++c) acc[s][c] = a * acc[s][c];
    }
  }
  out.store(acc, i);
}
