// Fused small-sequence attention forward/backward (north-star
// "transformer forward/backward" kernels; reference spec
// calibration/transformer_models.py:76-118 scaled_dot_product +
// per-sample-heads MHA, and the token-sequence variant in
// rl/transformer_networks.py).
//
// Shapes in this domain are SMALL: the supervised classifier attends
// over H<=8 "head tokens" (dh ~ 66), the RL calib/demix encoder over
// T = M+2 <= 32 sky tokens (dh = 16). So the whole attention —
// S = Q K^T / sqrt(dh), row softmax, O = A V — fits in LDS for one
// workgroup per (batch*head), ONE kernel launch instead of ~8 torch
// kernels (2 batched GEMMs + softmax + scales), and the backward is a
// second single launch. GEMM-shaped pieces run on v_mfma_f32_16x16x4_f32
// tiles (exact f32) with zero-padded LDS operands; row reductions use
// 16-lane DPP sums. Caps: T <= 32, dh <= 96 (python falls back to the
// torch composition beyond — no workload exceeds them).
//
// Layout: Q, K, V, O, A are (G, T, dh) / (G, T, T) contiguous fp32 with
// G = batch*heads.

#include "common.h"

#define ATTN_TMAX 32
#define ATTN_DMAX 96

// one wave per (b,h) group; dynamic LDS
extern "C" __global__ __launch_bounds__(64) void attn_fwd_kernel(
    const float* __restrict__ Q, const float* __restrict__ K,
    const float* __restrict__ V, float* __restrict__ O,
    float* __restrict__ A,        // (G, T, T) softmax output (for bwd)
    int G, int T, int dh) {
  extern __shared__ float lds[];
  const int Tp = (T + 15) & ~15;         // padded to 16-tile
  const int dp = ((dh + 3) & ~3) + 1;    // k-mult of 4, +1 bank pad
  float* qs = lds;                        // [Tp][dp]
  float* ks = qs + Tp * dp;               // [Tp][dp]
  float* vs = ks + Tp * dp;               // [Tp][dp]
  float* ss = vs + Tp * dp;               // [Tp][Tp+1]
  const int g = blockIdx.x;
  const int lane = threadIdx.x;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;
  const float scale = rsqrtf((float)dh);
  const long base = (long)g * T * dh;

  // stage Q/K/V zero-padded
  for (int i = lane; i < Tp * dp; i += WAVE) {
    const int r = i / dp, c = i % dp;
    const float qv = (r < T && c < dh) ? Q[base + (long)r * dh + c] : 0.f;
    const float kv = (r < T && c < dh) ? K[base + (long)r * dh + c] : 0.f;
    const float vv = (r < T && c < dh) ? V[base + (long)r * dh + c] : 0.f;
    qs[i] = qv;
    ks[i] = kv;
    vs[i] = vv;
  }
  __builtin_amdgcn_s_waitcnt(0);  // lgkm drain before fragment reads
  // (single wave: no barrier needed, but the compiler needs the ds waits)

  const int nt = Tp / 16;
  // S = Q K^T * scale  (NT product; k-loop over dh)
  for (int ti = 0; ti < nt; ++ti) {
    for (int tj = 0; tj < nt; ++tj) {
      f32x4 acc = (f32x4){0.f, 0.f, 0.f, 0.f};
      for (int k4 = 0; k4 < dp / 4; ++k4) {
        const int k = k4 * 4 + l4;
        const float a = qs[(ti * 16 + l15) * dp + k];
        const float b = ks[(tj * 16 + l15) * dp + k];
        acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
      }
#pragma unroll
      for (int r = 0; r < 4; ++r)
        ss[(ti * 16 + l4 * 4 + r) * (Tp + 1) + tj * 16 + l15] =
            acc[r] * scale;
    }
  }
  __builtin_amdgcn_s_waitcnt(0);

  // row softmax over j < T (rows strided across the wave)
  for (int i = lane; i < T; i += WAVE) {
    float m = -3.4e38f;
    for (int j = 0; j < T; ++j) m = fmaxf(m, ss[i * (Tp + 1) + j]);
    float s = 0.f;
    for (int j = 0; j < T; ++j) {
      const float e = __expf(ss[i * (Tp + 1) + j] - m);
      ss[i * (Tp + 1) + j] = e;
      s += e;
    }
    const float inv = 1.f / s;
    for (int j = 0; j < T; ++j) ss[i * (Tp + 1) + j] *= inv;
    for (int j = T; j < Tp; ++j) ss[i * (Tp + 1) + j] = 0.f;
  }
  for (int i = lane; i < Tp * Tp; i += WAVE) {  // zero pad rows for A@V
    const int r = i / Tp;
    if (r >= T) ss[r * (Tp + 1) + i % Tp] = 0.f;
  }
  __builtin_amdgcn_s_waitcnt(0);

  // write A (T,T) to global for backward
  for (int i = lane; i < T * T; i += WAVE)
    A[(long)g * T * T + i] = ss[(i / T) * (Tp + 1) + i % T];

  // O = A @ V  (NN product; k-loop over Tp)
  const int ndt = dp / 16 + ((dp % 16) ? 1 : 0);
  for (int ti = 0; ti < nt; ++ti) {
    for (int tj = 0; tj < ndt; ++tj) {
      f32x4 acc = (f32x4){0.f, 0.f, 0.f, 0.f};
      for (int k4 = 0; k4 < Tp / 4; ++k4) {
        const int k = k4 * 4 + l4;
        const float a = ss[(ti * 16 + l15) * (Tp + 1) + k];
        const int col = tj * 16 + l15;
        const float b = (col < dp) ? vs[k * dp + col] : 0.f;
        acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = ti * 16 + l4 * 4 + r;
        const int col = tj * 16 + l15;
        if (row < T && col < dh) O[base + (long)row * dh + col] = acc[r];
      }
    }
  }
}

// backward: given dO, A, Q, K, V -> dQ, dK, dV (scale folded into dQ/dK)
extern "C" __global__ __launch_bounds__(64) void attn_bwd_kernel(
    const float* __restrict__ DO, const float* __restrict__ A,
    const float* __restrict__ Q, const float* __restrict__ K,
    const float* __restrict__ V, float* __restrict__ DQ,
    float* __restrict__ DK, float* __restrict__ DV, int G, int T, int dh) {
  extern __shared__ float lds[];
  const int Tp = (T + 15) & ~15;
  const int dp = ((dh + 3) & ~3) + 1;
  float* dos = lds;                  // [Tp][dp]
  float* qs = dos + Tp * dp;         // [Tp][dp]
  float* ks = qs + Tp * dp;          // [Tp][dp]
  float* vs = ks + Tp * dp;          // [Tp][dp]
  float* as = vs + Tp * dp;          // [Tp][Tp+1] softmax A
  float* dss = as + Tp * (Tp + 1);   // [Tp][Tp+1] dA then dS
  float* rsum = dss + Tp * (Tp + 1); // [Tp]
  const int g = blockIdx.x;
  const int lane = threadIdx.x;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;
  const float scale = rsqrtf((float)dh);
  const long base = (long)g * T * dh;

  for (int i = lane; i < Tp * dp; i += WAVE) {
    const int r = i / dp, c = i % dp;
    const bool in = (r < T && c < dh);
    dos[i] = in ? DO[base + (long)r * dh + c] : 0.f;
    qs[i] = in ? Q[base + (long)r * dh + c] : 0.f;
    ks[i] = in ? K[base + (long)r * dh + c] : 0.f;
    vs[i] = in ? V[base + (long)r * dh + c] : 0.f;
  }
  for (int i = lane; i < Tp * (Tp + 1); i += WAVE) as[i] = 0.f;
  __builtin_amdgcn_s_waitcnt(0);
  for (int i = lane; i < T * T; i += WAVE)
    as[(i / T) * (Tp + 1) + i % T] = A[(long)g * T * T + i];
  __builtin_amdgcn_s_waitcnt(0);

  const int nt = Tp / 16;
  // dA = dO @ V^T (NT)
  for (int ti = 0; ti < nt; ++ti) {
    for (int tj = 0; tj < nt; ++tj) {
      f32x4 acc = (f32x4){0.f, 0.f, 0.f, 0.f};
      for (int k4 = 0; k4 < dp / 4; ++k4) {
        const int k = k4 * 4 + l4;
        const float a = dos[(ti * 16 + l15) * dp + k];
        const float b = vs[(tj * 16 + l15) * dp + k];
        acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
      }
#pragma unroll
      for (int r = 0; r < 4; ++r)
        dss[(ti * 16 + l4 * 4 + r) * (Tp + 1) + tj * 16 + l15] = acc[r];
    }
  }
  __builtin_amdgcn_s_waitcnt(0);
  // softmax backward: dS = A * (dA - rowsum(dA * A))
  for (int i = lane; i < T; i += WAVE) {
    float s = 0.f;
    for (int j = 0; j < T; ++j)
      s += dss[i * (Tp + 1) + j] * as[i * (Tp + 1) + j];
    rsum[i] = s;
  }
  __builtin_amdgcn_s_waitcnt(0);
  for (int i = lane; i < Tp * Tp; i += WAVE) {
    const int r = i / Tp, c = i % Tp;
    dss[r * (Tp + 1) + c] = (r < T)
        ? as[r * (Tp + 1) + c] * (dss[r * (Tp + 1) + c] - rsum[r])
        : 0.f;
  }
  __builtin_amdgcn_s_waitcnt(0);

  const int ndt = dp / 16 + ((dp % 16) ? 1 : 0);
  // dQ = scale * dS @ K (NN over k=Tp), dK = scale * dS^T @ Q,
  // dV = A^T @ dO
  for (int ti = 0; ti < nt; ++ti) {
    for (int tj = 0; tj < ndt; ++tj) {
      f32x4 aq = (f32x4){0.f, 0.f, 0.f, 0.f};
      f32x4 ak = (f32x4){0.f, 0.f, 0.f, 0.f};
      f32x4 av = (f32x4){0.f, 0.f, 0.f, 0.f};
      for (int k4 = 0; k4 < Tp / 4; ++k4) {
        const int k = k4 * 4 + l4;
        const int col = tj * 16 + l15;
        const float bK = (col < dp) ? ks[k * dp + col] : 0.f;
        const float bQ = (col < dp) ? qs[k * dp + col] : 0.f;
        const float bO = (col < dp) ? dos[k * dp + col] : 0.f;
        const float aS = dss[(ti * 16 + l15) * (Tp + 1) + k];   // dS[i][k]
        const float aST = dss[k * (Tp + 1) + ti * 16 + l15];    // dS[k][i]
        const float aAT = as[k * (Tp + 1) + ti * 16 + l15];     // A[k][i]
        aq = __builtin_amdgcn_mfma_f32_16x16x4f32(aS, bK, aq, 0, 0, 0);
        ak = __builtin_amdgcn_mfma_f32_16x16x4f32(aST, bQ, ak, 0, 0, 0);
        av = __builtin_amdgcn_mfma_f32_16x16x4f32(aAT, bO, av, 0, 0, 0);
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = ti * 16 + l4 * 4 + r;
        const int col = tj * 16 + l15;
        if (row < T && col < dh) {
          DQ[base + (long)row * dh + col] = aq[r] * scale;
          DK[base + (long)row * dh + col] = ak[r] * scale;
          DV[base + (long)row * dh + col] = av[r];
        }
      }
    }
  }
}
