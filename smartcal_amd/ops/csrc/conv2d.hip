// Direct (im2col-free) 2-D convolution kernels for the CNN agents'
// k5/s2/no-padding stacks (reference
// `calibration/calib_sac.py:99-104`: Conv2d 1→16→32→32, k5, s2 on
// 128×128 influence maps). Channel counts are tiny (≤32), so direct
// per-output accumulation is the right CDNA4 algorithm — an MFMA
// implicit GEMM would idle most of each 16×16 tile. Weights broadcast
// through L1/L2 (all threads of a block share cout), inputs are
// L2-resident between layers.
//
// Forward:  one thread per (b, cout, oy, ox); 25·Cin-term FMA chain
//           with front-loaded weight registers per (cin) slice.
// Backward: dx — one thread per input pixel, gathering the stride-2
//           compatible (cout, ky, kx) taps; dW/db — one block per
//           (cout, cin), grid-stride reduction over (b, oy, ox) with 25
//           per-thread accumulators combined through LDS.

#include "common.h"

#define KS 5
#define STRIDE 2

extern "C" __global__ __launch_bounds__(256) void conv2d_k5s2_fwd_kernel(
    const float* __restrict__ X,   // (B, Cin, H, W)
    const float* __restrict__ Wt,  // (Cout, Cin, 5, 5)
    const float* __restrict__ bias,// (Cout) or null
    float* __restrict__ Y,         // (B, Cout, OH, OW)
    int B, int Cin, int H, int W, int Cout, int OH, int OW) {
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long total = (long)B * Cout * OH * OW;
  if (idx >= total) return;
  const int ox = (int)(idx % OW);
  const int oy = (int)((idx / OW) % OH);
  const int co = (int)((idx / ((long)OW * OH)) % Cout);
  const int b = (int)(idx / ((long)OW * OH * Cout));
  const int iy0 = oy * STRIDE;
  const int ix0 = ox * STRIDE;

  float acc = bias ? bias[co] : 0.f;
  for (int ci = 0; ci < Cin; ++ci) {
    const float* xp = X + (((long)b * Cin + ci) * H + iy0) * W + ix0;
    const float* wp = Wt + (((long)co * Cin + ci) * KS) * KS;
    // front-load the 25 weights (L1-broadcast across the block), then a
    // 25-term FMA chain over the input patch
    float wr[KS * KS];
#pragma unroll
    for (int t = 0; t < KS * KS; ++t) wr[t] = wp[t];
#pragma unroll
    for (int ky = 0; ky < KS; ++ky) {
      const float* xr = xp + (long)ky * W;
#pragma unroll
      for (int kx = 0; kx < KS; ++kx)
        acc = fmaf(xr[kx], wr[ky * KS + kx], acc);
    }
  }
  Y[idx] = acc;
}

extern "C" __global__ __launch_bounds__(256) void conv2d_k5s2_dx_kernel(
    const float* __restrict__ DY,  // (B, Cout, OH, OW)
    const float* __restrict__ Wt,  // (Cout, Cin, 5, 5)
    float* __restrict__ DX,        // (B, Cin, H, W)
    int B, int Cin, int H, int W, int Cout, int OH, int OW) {
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long total = (long)B * Cin * H * W;
  if (idx >= total) return;
  const int ix = (int)(idx % W);
  const int iy = (int)((idx / W) % H);
  const int ci = (int)((idx / ((long)W * H)) % Cin);
  const int b = (int)(idx / ((long)W * H * Cin));

  float acc = 0.f;
  // taps: iy = oy*2 + ky  ⇒  oy = (iy-ky)/2 when divisible and in range
#pragma unroll
  for (int ky = 0; ky < KS; ++ky) {
    const int ry = iy - ky;
    if (ry < 0 || (ry & 1)) continue;
    const int oy = ry >> 1;
    if (oy >= OH) continue;
#pragma unroll
    for (int kx = 0; kx < KS; ++kx) {
      const int rx = ix - kx;
      if (rx < 0 || (rx & 1)) continue;
      const int ox = rx >> 1;
      if (ox >= OW) continue;
      for (int co = 0; co < Cout; ++co) {
        const float dz = DY[(((long)b * Cout + co) * OH + oy) * OW + ox];
        const float w = Wt[(((long)co * Cin + ci) * KS + ky) * KS + kx];
        acc = fmaf(dz, w, acc);
      }
    }
  }
  DX[idx] = acc;
}

extern "C" __global__ __launch_bounds__(256) void conv2d_k5s2_dw_kernel(
    const float* __restrict__ DY,  // (B, Cout, OH, OW)
    const float* __restrict__ X,   // (B, Cin, H, W)
    float* __restrict__ DW,        // (Cout, Cin, 5, 5)
    float* __restrict__ DB,        // (Cout) — written by blocks with ci==0
    int B, int Cin, int H, int W, int Cout, int OH, int OW) {
  // block = one (cout, cin) pair; threads grid-stride over (b, oy, ox)
  // keeping 25 partial dW accumulators (+1 for db) in registers
  const int co = blockIdx.x / Cin;
  const int ci = blockIdx.x % Cin;
  const int tid = threadIdx.x;
  float acc[KS * KS];
#pragma unroll
  for (int t = 0; t < KS * KS; ++t) acc[t] = 0.f;
  float accb = 0.f;
  const long total = (long)B * OH * OW;
  for (long i = tid; i < total; i += blockDim.x) {
    const int ox = (int)(i % OW);
    const int oy = (int)((i / OW) % OH);
    const int b = (int)(i / ((long)OW * OH));
    const float dz = DY[(((long)b * Cout + co) * OH + oy) * OW + ox];
    accb += dz;
    const float* xp = X + (((long)b * Cin + ci) * H + oy * STRIDE) * W
                      + ox * STRIDE;
#pragma unroll
    for (int ky = 0; ky < KS; ++ky) {
      const float* xr = xp + (long)ky * W;
#pragma unroll
      for (int kx = 0; kx < KS; ++kx)
        acc[ky * KS + kx] = fmaf(dz, xr[kx], acc[ky * KS + kx]);
    }
  }
  // block reduction through LDS, one weight tap at a time
  __shared__ float red[256];
  for (int t = 0; t < KS * KS; ++t) {
    red[tid] = acc[t];
    __syncthreads();
    for (int off = 128; off > 0; off >>= 1) {
      if (tid < off) red[tid] += red[tid + off];
      __syncthreads();
    }
    if (tid == 0)
      DW[(((long)co * Cin + ci) * KS) * KS + t] = red[0];
    __syncthreads();
  }
  red[tid] = accb;
  __syncthreads();
  for (int off = 128; off > 0; off >>= 1) {
    if (tid < off) red[tid] += red[tid + off];
    __syncthreads();
  }
  if (tid == 0 && ci == 0) DB[co] = red[0];
}
