// quda_amd HIP device common: complex math, chunked-SoA field accessors,
// checkerboard index helpers. MI355X (gfx950) only — wave64, 16-byte
// vectorized global access (cdna_hip_programming.md G2/G13).
//
// Layouts match quda_amd/fields/layout.py:
//   spinor: [n_chunk][V_cb][W] reals, comp = (s*3+c)*2+reim, W*sizeof = 16B
//   gauge : [mu][parity][n_chunk][V_cb][2] complex-pair chunks
//   clover: [parity][n_chunk][V_cb][W], 72 reals packed (fields/clover.py)
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_fp16.h>
#include <hip/hip_fp8.h>

// ---------------------------------------------------------------------------
// complex
// ---------------------------------------------------------------------------
template <typename T>
struct cplx {
  T re, im;
  __device__ __forceinline__ cplx() {}
  __device__ __forceinline__ cplx(T r, T i) : re(r), im(i) {}
  __device__ __forceinline__ cplx operator+(const cplx &o) const { return {re + o.re, im + o.im}; }
  __device__ __forceinline__ cplx operator-(const cplx &o) const { return {re - o.re, im - o.im}; }
  __device__ __forceinline__ cplx operator*(const cplx &o) const {
    return {re * o.re - im * o.im, re * o.im + im * o.re};
  }
  __device__ __forceinline__ cplx &operator+=(const cplx &o) { re += o.re; im += o.im; return *this; }
};

template <typename T>
__device__ __forceinline__ cplx<T> operator*(T a, const cplx<T> &z) { return {a * z.re, a * z.im}; }
template <typename T>
__device__ __forceinline__ cplx<T> imul(const cplx<T> &z) { return {-z.im, z.re}; }   // i*z
template <typename T>
__device__ __forceinline__ cplx<T> mimul(const cplx<T> &z) { return {z.im, -z.re}; }  // -i*z
template <typename T>
__device__ __forceinline__ cplx<T> neg(const cplx<T> &z) { return {-z.re, -z.im}; }
template <typename T>
__device__ __forceinline__ cplx<T> conj(const cplx<T> &z) { return {z.re, -z.im}; }
// fused a*b + c
template <typename T>
__device__ __forceinline__ cplx<T> cfma(const cplx<T> &a, const cplx<T> &b, const cplx<T> &c) {
  return {fma(a.re, b.re, fma(-a.im, b.im, c.re)), fma(a.re, b.im, fma(a.im, b.re, c.im))};
}
// conj(a)*b + c
template <typename T>
__device__ __forceinline__ cplx<T> cfma_conj(const cplx<T> &a, const cplx<T> &b, const cplx<T> &c) {
  return {fma(a.re, b.re, fma(a.im, b.im, c.re)), fma(a.re, b.im, fma(-a.im, b.re, c.im))};
}

// ---------------------------------------------------------------------------
// storage traits: Store = memory type, Real = compute type
// ---------------------------------------------------------------------------
// distinct 1-byte wrapper so overloads can route fp8-e4m3 (OCP, gfx950
// hardware-converted) storage through the block-float decode/encode —
// the "quarter" precision (ref: color_spinor_field_order.h:1426 FLOAT8 /
// quarter fixed-point, re-based on true fp8 since CDNA4 has native
// e4m3 converts)
struct fp8s {
  unsigned char b;
};

struct PrecDouble  { using Store = double; using Real = double; static constexpr int W = 2;  static constexpr bool has_norm = false; };
struct PrecSingle  { using Store = float;  using Real = float;  static constexpr int W = 4;  static constexpr bool has_norm = false; };
struct PrecHalf    { using Store = __half; using Real = float;  static constexpr int W = 8;  static constexpr bool has_norm = true;  };
struct PrecQuarter { using Store = fp8s;   using Real = float;  static constexpr int W = 16; static constexpr bool has_norm = true;  };

template <typename R, typename S>
__device__ __forceinline__ R qa_tor(const S &v) { return (R)v; }
template <typename R>
__device__ __forceinline__ R qa_tor(const fp8s &v) {
  return (R)__half(__hip_cvt_fp8_to_halfraw(v.b, __HIP_E4M3));
}

// decode two adjacent stored values (one complex) at once; the fp8
// specialization uses the CDNA4 packed converter v_cvt_pk_f32_fp8
// (gfx950 native OCP e4m3) — one VALU op for the pair instead of two
// byte-wise __hip_cvt chains (the quarter dslash is decode-VALU bound)
template <typename R, typename S>
__device__ __forceinline__ void qa_tor_pair(const S *p, R &a, R &b) {
  a = qa_tor<R>(p[0]);
  b = qa_tor<R>(p[1]);
}
template <>
__device__ __forceinline__ void qa_tor_pair<float, fp8s>(const fp8s *p,
                                                         float &a, float &b) {
  unsigned short u = (unsigned short)((p[1].b << 8) | p[0].b);
  auto v = __builtin_amdgcn_cvt_pk_f32_fp8(u, false);
  a = v[0];
  b = v[1];
}

template <typename S, typename R>
__device__ __forceinline__ S qa_tos(const R &v) { return (S)v; }
template <>
__device__ __forceinline__ fp8s qa_tos<fp8s, float>(const float &v) {
  return {__hip_cvt_float_to_fp8(v, __HIP_SATFINITE, __HIP_E4M3)};
}

__device__ __forceinline__ void qa_sincospi(float x, float *s, float *c) {
  sincospif(x, s, c);  // sin(pi x), cos(pi x)
}
__device__ __forceinline__ void qa_sincospi(double x, double *s, double *c) {
  sincospi(x, s, c);
}

// 16-byte opaque chunk for vector loads
struct alignas(16) chunk16 { unsigned int u[4]; };

template <typename S, int W>
__device__ __forceinline__ void load_chunk(const S *p, S out[W]) {
  if constexpr (W * sizeof(S) == 16) {
    *reinterpret_cast<chunk16 *>(out) = *reinterpret_cast<const chunk16 *>(p);
  } else if constexpr (W * sizeof(S) == 8) {
    *reinterpret_cast<unsigned long long *>(out) = *reinterpret_cast<const unsigned long long *>(p);
  } else {
    *reinterpret_cast<unsigned int *>(out) = *reinterpret_cast<const unsigned int *>(p);
  }
}

template <typename S, int W>
__device__ __forceinline__ void store_chunk(S *p, const S in[W]) {
  if constexpr (W * sizeof(S) == 16) {
    *reinterpret_cast<chunk16 *>(p) = *reinterpret_cast<const chunk16 *>(in);
  } else if constexpr (W * sizeof(S) == 8) {
    *reinterpret_cast<unsigned long long *>(p) = *reinterpret_cast<const unsigned long long *>(in);
  } else {
    *reinterpret_cast<unsigned int *>(p) = *reinterpret_cast<const unsigned int *>(in);
  }
}

// ---------------------------------------------------------------------------
// spinor accessor: NCOMP reals/site in [NCOMP/CW][V][CW]; CW is the widest
// chunk (<= Prec::W) dividing NCOMP (24 -> 16B chunks everywhere; staggered
// 6 -> one complex per chunk, the reference's staggered FloatN choice)
// ---------------------------------------------------------------------------
template <int NC, int W>
struct chunk_w {
  static constexpr int value = (NC % W == 0) ? W : chunk_w<NC, W / 2>::value;
};
template <int NC>
struct chunk_w<NC, 1> {
  static constexpr int value = 1;
};

template <typename Prec, int NCOMP = 24>
struct SpinorAcc {
  using S = typename Prec::Store;
  using R = typename Prec::Real;
  static constexpr int W = chunk_w<NCOMP, Prec::W>::value;
  static constexpr int NCH = NCOMP / W;
  static constexpr int NCPLX = NCOMP / 2;
  S *data;
  float *norm;  // only for half
  long V;       // cb volume (chunk stride); fields span npar * V sites

  // global site g in [0, npar*V): parity block p = g/V without a divide
  __device__ __forceinline__ long chunk_base(long g) const {
    long p = (g >= V) ? 1 : 0;
    return (p * NCH * V + (g - p * V)) * W;
  }

  __device__ __forceinline__ void load_v(cplx<R> out[NCPLX], long g) const {
    S tmp[NCOMP];
    long base = chunk_base(g);
#pragma unroll
    for (int ch = 0; ch < NCH; ++ch)
      load_chunk<S, W>(data + base + (long)ch * V * W, tmp + ch * W);
    R scale = (R)1;
    if constexpr (Prec::has_norm) scale = norm[g];
#pragma unroll
    for (int k = 0; k < NCPLX; ++k) {
      R re, im;
      qa_tor_pair<R, S>(tmp + 2 * k, re, im);
      out[k] = {scale * re, scale * im};
    }
  }

  __device__ __forceinline__ void store_v(const cplx<R> in[NCPLX], long g) const {
    S tmp[NCOMP];
    long base = chunk_base(g);
    if constexpr (Prec::has_norm) {
      R m = (R)0;
#pragma unroll
      for (int k = 0; k < NCPLX; ++k)
        m = fmax(m, fmax(fabs(in[k].re), fabs(in[k].im)));
      norm[g] = m;
      R inv = m > (R)0 ? (R)1 / m : (R)0;
#pragma unroll
      for (int k = 0; k < NCPLX; ++k) {
        tmp[2 * k] = qa_tos<S>(in[k].re * inv);
        tmp[2 * k + 1] = qa_tos<S>(in[k].im * inv);
      }
    } else {
#pragma unroll
      for (int k = 0; k < NCPLX; ++k) {
        tmp[2 * k] = (S)in[k].re;
        tmp[2 * k + 1] = (S)in[k].im;
      }
    }
#pragma unroll
    for (int ch = 0; ch < NCH; ++ch)
      store_chunk<S, W>(data + base + (long)ch * V * W, tmp + ch * W);
  }

  // [4][3]-shaped views for the Wilson kernels (NCOMP == 24 only)
  __device__ __forceinline__ void load_g(cplx<R> (&out)[4][3], long g) const {
    static_assert(NCOMP == 24);
    load_v(reinterpret_cast<cplx<R> *>(out), g);
  }
  __device__ __forceinline__ void store_g(const cplx<R> (&in)[4][3], long g) const {
    static_assert(NCOMP == 24);
    store_v(reinterpret_cast<const cplx<R> *>(in), g);
  }

  // single-parity (dslash) aliases: site index i in [0, V)
  __device__ __forceinline__ void load(cplx<R> (&out)[4][3], long i) const { load_g(out, i); }
  __device__ __forceinline__ void store(const cplx<R> (&in)[4][3], long i) const { store_g(in, i); }
};

// staggered (nSpin=1) spinor: 6 reals/site, one complex color per chunk
template <typename Prec>
using StagAcc = SpinorAcc<Prec, 6>;

// ---------------------------------------------------------------------------
// gauge accessor — "stencil" layout (quda_amd/fields/gauge.py):
// per parity, per site: 8 links (slots Q = 0-3 fwd U_mu(x), 4-7 bwd
// U_mu(x-mu), pre-shifted) packed contiguously, 16-byte chunked.
// recon-12: row2 = conj(row0 x row1) (ref gauge_field_order.h:2369).
// Every load is site-local; the slot Q is compile-time so the chunk range
// covering elements [Q*RECON, (Q+1)*RECON) folds to constants.
// ---------------------------------------------------------------------------
template <typename Prec, int RECON>
struct GaugeAcc {
  using S = typename Prec::Store;
  using R = typename Prec::Real;
  static constexpr int W = Prec::W;
  static constexpr int NCH = (8 * RECON) / W;  // chunks per site
  const S *data;   // parity-adjusted base: [NCH][V][W]
  const S *other;  // OPPOSITE-parity base (bwd links read from the -mu
                   // neighbor's fwd slot live there: x-mu flips parity)
  long V;

  // split raw-load / decode pair (RECON 12/18): lets a kernel issue the
  // chunk loads of many links early (e.g. behind an LDS fill) and decode
  // at use. Chunk count: recon-12 at W=8 always spans 2 chunks.
  template <int Q>
  static constexpr int raw_nc() {
    return (Q * RECON + RECON - 1) / W - (Q * RECON) / W + 1;
  }
  template <int Q>
  __device__ __forceinline__ void load_raw(const S *base, S *out,
                                           long i) const {
    constexpr int c0 = (Q * RECON) / W;
#pragma unroll
    for (int c = 0; c < raw_nc<Q>(); ++c)
      load_chunk<S, W>(base + ((long)(c0 + c) * V + i) * W, out + c * W);
  }
  template <int Q>
  __device__ __forceinline__ void decode_raw(const S *in,
                                             cplx<R> u[3][3]) const {
    static_assert(RECON == 12 || RECON == 18, "raw split: plain codecs only");
    constexpr int off = Q * RECON - ((Q * RECON) / W) * W;
#pragma unroll
    for (int k = 0; k < RECON / 2; ++k)
      u[k / 3][k % 3] = {qa_tor<R>(in[off + 2 * k]),
                        qa_tor<R>(in[off + 2 * k + 1])};
    if constexpr (RECON == 12) {
#pragma unroll
      for (int c = 0; c < 3; ++c) {
        int a = (c + 1) % 3, b = (c + 2) % 3;
        u[2][c] = conj(u[0][a] * u[1][b] - u[0][b] * u[1][a]);
      }
    }
  }

  template <int Q>
  __device__ __forceinline__ void load_base(const S *base, cplx<R> u[3][3],
                                            long i) const {
    constexpr int e0 = Q * RECON;
    constexpr int c0 = e0 / W;
    constexpr int c1 = (e0 + RECON - 1) / W;
    constexpr int NC = c1 - c0 + 1;
    constexpr int off = e0 - c0 * W;
    S tmp[NC * W];
#pragma unroll
    for (int c = 0; c < NC; ++c)
      load_chunk<S, W>(base + ((long)(c0 + c) * V + i) * W, tmp + c * W);
    if constexpr (RECON == 8) {
      // arXiv:0911.3191 codec in the row-permuted {{b},{a},{-c}} form
      // (ref gauge_field_order.h Reconstruct<8>; u0 = 1 — anisotropy and
      // boundary phases force reconstruct 'none' at load time)
      R st[8];
#pragma unroll
      for (int k = 0; k < 8; ++k) st[k] = qa_tor<R>(tmp[off + k]);
      cplx<R> b2{st[2], st[3]}, b3{st[4], st[5]}, a1{st[6], st[7]};
      R row_sum = b2.re * b2.re + b2.im * b2.im + b3.re * b3.re + b3.im * b3.im;
      R rinv = (R)1 / row_sum;
      R d1 = (R)1 - row_sum;
      R b1m = d1 > (R)0 ? sqrt(d1) : (R)0;
      R sn, cs;
      qa_sincospi(st[0], &sn, &cs);
      cplx<R> b1{cs * b1m, sn * b1m};
      R col_sum = b1.re * b1.re + b1.im * b1.im + a1.re * a1.re + a1.im * a1.im;
      R d2 = (R)1 - col_sum;
      R cm = d2 > (R)0 ? sqrt(d2) : (R)0;
      qa_sincospi(st[1], &sn, &cs);
      cplx<R> mc1{cs * cm, sn * cm};  // -c1
      cplx<R> A = conj(b1) * a1;
      cplx<R> a2 = neg(conj(mc1) * conj(b3) + A * b2);
      a2 = {a2.re * rinv, a2.im * rinv};
      cplx<R> a3 = conj(mc1) * conj(b2) - A * b3;
      a3 = {a3.re * rinv, a3.im * rinv};
      cplx<R> B = conj(b1) * mc1;
      cplx<R> c2 = neg(conj(a1) * conj(b3) - B * b2);  // -(mc2)
      c2 = {c2.re * rinv, c2.im * rinv};
      cplx<R> c3 = conj(a1) * conj(b2) + B * b3;       // -(mc3)
      c3 = {c3.re * rinv, c3.im * rinv};
      u[0][0] = a1;
      u[0][1] = a2;
      u[0][2] = a3;
      u[1][0] = b1;
      u[1][1] = b2;
      u[1][2] = b3;
      u[2][0] = neg(mc1);
      u[2][1] = c2;
      u[2][2] = c3;
      return;
    }
#pragma unroll
    for (int k = 0; k < RECON / 2; ++k) {
      R re, im;
      qa_tor_pair<R, S>(tmp + off + 2 * k, re, im);
      u[k / 3][k % 3] = {re, im};
    }
    if constexpr (RECON == 12) {
#pragma unroll
      for (int c = 0; c < 3; ++c) {
        int a = (c + 1) % 3, b = (c + 2) % 3;
        u[2][c] = conj(u[0][a] * u[1][b] - u[0][b] * u[1][a]);
      }
    }
  }

  template <int Q>
  __device__ __forceinline__ void load(cplx<R> u[3][3], long i) const {
    load_base<Q>(data, u, i);
  }
  // fwd slot Q at site i of the OTHER parity block
  template <int Q>
  __device__ __forceinline__ void load_o(cplx<R> u[3][3], long i) const {
    load_base<Q>(other, u, i);
  }
};

// U * h (3x3 times 2x3 half-spinor, h indexed [spin][color])
template <typename R>
__device__ __forceinline__ void su3_mul_half(cplx<R> out[2][3], const cplx<R> u[3][3],
                                             const cplx<R> h[2][3]) {
#pragma unroll
  for (int s = 0; s < 2; ++s)
#pragma unroll
    for (int r = 0; r < 3; ++r) {
      cplx<R> acc = u[r][0] * h[s][0];
      acc = cfma(u[r][1], h[s][1], acc);
      acc = cfma(u[r][2], h[s][2], acc);
      out[s][r] = acc;
    }
}

// U^dag * h
template <typename R>
__device__ __forceinline__ void su3_dagmul_half(cplx<R> out[2][3], const cplx<R> u[3][3],
                                                const cplx<R> h[2][3]) {
#pragma unroll
  for (int s = 0; s < 2; ++s)
#pragma unroll
    for (int r = 0; r < 3; ++r) {
      cplx<R> acc = cfma_conj(u[0][r], h[s][0], cplx<R>((R)0, (R)0));
      acc = cfma_conj(u[1][r], h[s][1], acc);
      acc = cfma_conj(u[2][r], h[s][2], acc);
      out[s][r] = acc;
    }
}

// ---------------------------------------------------------------------------
// checkerboard coordinates (ref: include/index_helper.cuh coordsFromIndex)
// ---------------------------------------------------------------------------
struct LatDims {
  int X[4];
  int parity_offset;
  long Vcb;
};

__device__ __forceinline__ void coords_from_cb(int x[4], long i, const LatDims &d, int parity) {
  int X0h = d.X[0] >> 1;
  x[0] = (int)(i % X0h);
  long j = i / X0h;
  x[1] = (int)(j % d.X[1]);
  j /= d.X[1];
  x[2] = (int)(j % d.X[2]);
  x[3] = (int)(j / d.X[2]);
  int odd = (x[1] + x[2] + x[3] + parity + d.parity_offset) & 1;
  x[0] = 2 * x[0] + odd;
}

__device__ __forceinline__ long cb_from_coords(const int x[4], const LatDims &d) {
  long lex = (((long)x[3] * d.X[2] + x[2]) * d.X[1] + x[1]) * d.X[0] + x[0];
  return lex >> 1;
}

// neighbor cb index with periodic wrap in local lattice; dir = +1/-1
__device__ __forceinline__ long neighbor_cb(const int x[4], int mu, int dir, const LatDims &d) {
  int y[4] = {x[0], x[1], x[2], x[3]};
  int v = y[mu] + dir;
  if (v >= d.X[mu]) v -= d.X[mu];
  if (v < 0) v += d.X[mu];
  y[mu] = v;
  return cb_from_coords(y, d);
}

// kernel roles for the comm-overlap split (role of the reference's
// INTERIOR/EXTERIOR kernel_type, include/dslash_helper.cuh):
//   LOCAL    : no partitioned dims — pure local stencil (zero overhead)
//   FUSED    : ghost-aware single pass (blocking-comms policy)
//   INTERIOR : skip hops crossing a partitioned boundary; non-affine
//              epilogues on boundary sites are deferred to EXTERIOR
// The *_exterior kernels add the ghost hops and complete deferred
// epilogues; one owner thread per boundary site.
enum DslashKT { KT_LOCAL = 0, KT_FUSED = 1, KT_INTERIOR = 2 };

#define HIP_CHECK(cmd)                                                       \
  do {                                                                         \
    hipError_t e = (cmd);                                                      \
    if (e != hipSuccess) {                                                     \
      printf("HIP error %s at %s:%d\n", hipGetErrorString(e), __FILE__, __LINE__); \
    }                                                                          \
  } while (0)
