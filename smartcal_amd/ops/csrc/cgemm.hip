// Batched complex64 GEMM on f32 MFMA — the influence-core hot op.
//
// dsolutions_r (radio/hessian.py, reference calibration_tools.py:827-875)
// computes dJ = (H+eps)^-1 @ AdV as a batched complex GEMM
// (K dirs x 8 perturbations, 4N x 4N @ 4N x B; N=62 => 248x248 @ 248x1891).
// rocBLAS runs these complex shapes at ~1.1 TF/s (62% of the demix env
// step, profiles/r3_demix_step_kernel_stats.txt). This kernel computes the
// complex product as 4 real MFMA streams (re: ArBr - AiBi, im: ArBi + AiBr)
// on v_mfma_f32_16x16x4_f32 with both operands LDS-staged in split
// re/im planes.
//
// Layout: A (KA, M, K), B (G, K, N), C (G, M, N) complex64 row-major
// interleaved; batch g uses A[g / (G/KA)] (the Ainv-broadcast-over-8-dirs
// pattern). Block = 4 waves = 64 output rows x 16 cols per workgroup.

#include "common.h"

#define CBK 32   // K-chunk staged per iteration

extern "C" __global__ __launch_bounds__(256) void cgemm_nn_bcast_kernel(
    const float* __restrict__ A,   // (KA, M, K) interleaved re,im
    const float* __restrict__ B,   // (G, K, N)
    float* __restrict__ C,         // (G, M, N)
    int M, int K, int N, int G, int rep) {
  // LDS: A planes [64][CBK+1], B planes [CBK][17]
  __shared__ float ar_s[64 * (CBK + 1)];
  __shared__ float ai_s[64 * (CBK + 1)];
  __shared__ float br_s[CBK * 17];
  __shared__ float bi_s[CBK * 17];

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;
  const int row0 = blockIdx.x * 64;          // block's first output row
  const int col0 = blockIdx.y * 16;
  const int g = blockIdx.z;
  const long abase = (long)(g / rep) * M * K * 2;
  const long bbase = (long)g * K * N * 2;
  const long cbase = (long)g * M * N * 2;

  f32x4 acc_re = (f32x4){0.f, 0.f, 0.f, 0.f};
  f32x4 acc_im = (f32x4){0.f, 0.f, 0.f, 0.f};

  for (int kk = 0; kk < K; kk += CBK) {
    // stage A rows [row0, row0+64) x k-chunk — 64*32 cplx, 8 cplx/thread
    for (int i = tid; i < 64 * CBK; i += 256) {
      const int r = i / CBK, c = i % CBK;
      const int gr = row0 + r, gk = kk + c;
      float re = 0.f, im = 0.f;
      if (gr < M && gk < K) {
        const long o = abase + ((long)gr * K + gk) * 2;
        re = A[o];
        im = A[o + 1];
      }
      ar_s[r * (CBK + 1) + c] = re;
      ai_s[r * (CBK + 1) + c] = im;
    }
    // stage B k-chunk x cols [col0, col0+16) — 32*16 cplx, 2/thread
    for (int i = tid; i < CBK * 16; i += 256) {
      const int r = i / 16, c = i % 16;
      const int gk = kk + r, gc = col0 + c;
      float re = 0.f, im = 0.f;
      if (gk < K && gc < N) {
        const long o = bbase + ((long)gk * N + gc) * 2;
        re = B[o];
        im = B[o + 1];
      }
      br_s[r * 17 + c] = re;
      bi_s[r * 17 + c] = im;
    }
    __syncthreads();

#pragma unroll
    for (int k4 = 0; k4 < CBK / 4; ++k4) {
      const int k = k4 * 4 + l4;
      const float a_re = ar_s[(wave * 16 + l15) * (CBK + 1) + k];
      const float a_im = ai_s[(wave * 16 + l15) * (CBK + 1) + k];
      const float b_re = br_s[k * 17 + l15];
      const float b_im = bi_s[k * 17 + l15];
      acc_re = __builtin_amdgcn_mfma_f32_16x16x4f32(a_re, b_re, acc_re,
                                                    0, 0, 0);
      acc_re = __builtin_amdgcn_mfma_f32_16x16x4f32(-a_im, b_im, acc_re,
                                                    0, 0, 0);
      acc_im = __builtin_amdgcn_mfma_f32_16x16x4f32(a_re, b_im, acc_im,
                                                    0, 0, 0);
      acc_im = __builtin_amdgcn_mfma_f32_16x16x4f32(a_im, b_re, acc_im,
                                                    0, 0, 0);
    }
    __syncthreads();
  }

#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int orow = row0 + wave * 16 + l4 * 4 + r;
    const int ocol = col0 + l15;
    if (orow < M && ocol < N) {
      const long o = cbase + ((long)orow * N + ocol) * 2;
      C[o] = acc_re[r];
      C[o + 1] = acc_im[r];
    }
  }
}
