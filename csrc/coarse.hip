// Coarse-grid dslash on MFMA matrix cores (gfx950).
// (role of reference kernels/dslash_coarse_mma.cuh:755 — redesigned for
//  CDNA4: the coarse op out[a] = X[a] c[a] + sum_d Y[d][a] c[nbr(a,d)] is
//  9 complex [Nc x Nc] x [Nc x NR] GEMMs per site; one wave computes a
//  16-row x 16-rhs complex tile with v_mfma_f32_16x16x4_f32 pairs
//  (exact f32: bitwise an fmaf chain, cdna_hip_programming.md §3), fusing
//  ALL 9 direction matrices into one launch instead of 9 einsum/rocBLAS
//  dispatches + 8 neighbor gather copies.)
//
// Layouts (complex64 interleaved re,im):
//   mats  [9][Na][Nc][Nc]   m=0 is X, m=1..8 is Y[m-1]
//   nbr9  [Na][9] int64     source site per matrix (m=0 -> a itself);
//                           entries >= Na index appended ghost rows of c
//   c     [Nc_ext][Nc][NR]  RHS block (Nc_ext = Na + ghost sites)
//   out   [Na][Nc][NR]
// Nc must be a multiple of 16; NR <= 16 (lanes with j >= NR are padded).
#include <hip/hip_runtime.h>

#include "launchers.h"

using f32x4 = __attribute__((ext_vector_type(4))) float;

struct c64 {
  float re, im;
};

__global__ __launch_bounds__(256) void k_coarse_dslash_mfma(
    const c64 *__restrict__ mats, const long *__restrict__ nbr9,
    const c64 *__restrict__ c, c64 *__restrict__ out, long Na, int Nc,
    int NR) {
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int ntile = Nc >> 4;  // row tiles per site
  const long tile = (long)blockIdx.x * 4 + wave;
  const long a = tile / ntile;
  if (a >= Na) return;
  const int r0 = (int)(tile % ntile) << 4;
  // fragment maps (cdna_hip_programming.md §3, 16x16x4 f32):
  //   A: lane l supplies A[i = l&15][k = l>>4]
  //   B: lane l supplies B[k = l>>4][j = l&15]
  //   D: lane l holds rows (l>>4)*4+q (q=0..3) of column l&15
  const int i = lane & 15;
  const int kq = lane >> 4;
  f32x4 Dr = {0.f, 0.f, 0.f, 0.f};
  f32x4 Di = {0.f, 0.f, 0.f, 0.f};
  const long nbr_base = a * 9;
#pragma unroll 1
  for (int m = 0; m < 9; ++m) {
    const long src = nbr9[nbr_base + m];
    const c64 *__restrict__ M =
        mats + (((long)m * Na + a) * Nc + r0) * Nc;
    const c64 *__restrict__ B = c + src * (long)Nc * NR;
    for (int k0 = 0; k0 < Nc; k0 += 4) {
      c64 av = M[(long)i * Nc + k0 + kq];
      c64 bv = {0.f, 0.f};
      if (i < NR) bv = B[(long)(k0 + kq) * NR + i];
      // complex tile product via 4 real MFMAs:
      //   Dr += Ar Br - Ai Bi ;  Di += Ar Bi + Ai Br
      Dr = __builtin_amdgcn_mfma_f32_16x16x4f32(av.re, bv.re, Dr, 0, 0, 0);
      Dr = __builtin_amdgcn_mfma_f32_16x16x4f32(-av.im, bv.im, Dr, 0, 0, 0);
      Di = __builtin_amdgcn_mfma_f32_16x16x4f32(av.re, bv.im, Di, 0, 0, 0);
      Di = __builtin_amdgcn_mfma_f32_16x16x4f32(av.im, bv.re, Di, 0, 0, 0);
    }
  }
  const int col = lane & 15;
  const int rq = lane >> 4;
  if (col < NR) {
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      int r = r0 + rq * 4 + q;
      out[((long)a * Nc + r) * NR + col] = {Dr[q], Di[q]};
    }
  }
}

void launch_coarse_dslash_mfma(const CoarseMfmaCall &cc, hipStream_t st) {
  long ntiles = cc.Na * (cc.Nc / 16);
  int grid = (int)((ntiles + 3) / 4);
  hipLaunchKernelGGL(k_coarse_dslash_mfma, dim3(grid), dim3(256), 0, st,
                     (const c64 *)cc.mats, (const long *)cc.nbr9,
                     (const c64 *)cc.c, (c64 *)cc.out, cc.Na, cc.Nc, cc.NR);
}
