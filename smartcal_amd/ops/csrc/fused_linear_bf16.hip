// bf16-compute fused Linear(+LayerNorm)(+activation) + backward GEMMs.
//
// BASELINE config 2 is "elastic-net TD3 + prioritized replay, bf16". The
// bf16 design keeps fp32 MASTER weights, fp32 interfaces and fp32
// LayerNorm statistics/epilogue — only the GEMM multiplies run on the
// bf16 matrix pipe (v_mfma_f32_16x16x32_bf16, fp32 accumulate, 8x the
// K-depth of the f32 MFMA): operands are cast to bf16 while staging
// into LDS/registers, so activations and weights never round-trip
// through a bf16 tensor in HBM. Numerics: one bf16 rounding per operand
// per GEMM, fp32 everywhere else — mixed precision in the
// master-weights sense, matching how the parity test defines "bf16".
//
// Fragment mapping (validated on-box by gpurun_scripts/mfma_bf16_probe):
// C/D is the dtype-independent map col=lane&15, row=(lane>>4)*4+reg; for
// A/B we use the contiguous per-lane k-run k = (lane>>4)*8 + r, r<8 —
// the MFMA dot product is invariant under any k-permutation applied to
// BOTH operands, so a consistent map is sufficient (probe's random-GEMM
// check passes at bf16 tolerance for this map).

#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;

#define NWAVES 8
#define NT_MAX 12          // N <= 16*NWAVES*NT_MAX = 1536
#define BK 64              // K-tile staged in LDS
#define XPAD 8             // bf16 row pad (16 B) for conflict-free b128 reads
#define WCH 4              // float4 chunks per lane per W subtile

extern "C" __global__ __launch_bounds__(512) void fused_linear_bf16_fwd_kernel(
    const float* __restrict__ X,    // (B, K) fp32
    const float* __restrict__ W,    // (N, K) fp32 row-major (master)
    const float* __restrict__ bias, // (N) or null
    const float* __restrict__ gamma,// (N) or null (=> no LN)
    const float* __restrict__ beta, // (N)
    float* __restrict__ Y,          // (B, N) fp32
    float* __restrict__ ZHAT,       // (B, N) (LN only)
    float* __restrict__ RSTD,       // (B,)   (LN only)
    int B, int K, int N, int act, int with_ln) {
  __shared__ __bf16 xs[16 * (BK + XPAD)];
  __shared__ __bf16 ws[NWAVES * 16 * (BK + XPAD)];
  __shared__ float rowstat[NWAVES * 16 * 2];
  __shared__ float rowmv[16 * 2];

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;          // 0..3
  const int row0 = blockIdx.x * 16;
  const int ntiles = (N + 15) >> 4;
  const int XS = BK + XPAD;

  const int wc_row[WCH] = {(lane + 0 * WAVE) / (BK / 4),
                           (lane + 1 * WAVE) / (BK / 4),
                           (lane + 2 * WAVE) / (BK / 4),
                           (lane + 3 * WAVE) / (BK / 4)};
  const int wc_k[WCH] = {((lane + 0 * WAVE) % (BK / 4)) * 4,
                         ((lane + 1 * WAVE) % (BK / 4)) * 4,
                         ((lane + 2 * WAVE) % (BK / 4)) * 4,
                         ((lane + 3 * WAVE) % (BK / 4)) * 4};

  f32x4 acc[NT_MAX];
#pragma unroll
  for (int t = 0; t < NT_MAX; ++t) acc[t] = (f32x4){0.f, 0.f, 0.f, 0.f};

  __bf16* wsw = ws + wave * 16 * XS;

  for (int kk = 0; kk < K; kk += BK) {
    // 4 elements per thread, one packed 8-byte LDS store each
    for (int idx = tid * 4; idx < 16 * BK; idx += 512 * 4) {
      const int r = idx / BK, c = idx % BK;
      const int gr = row0 + r;
      bf16x4 p;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int gc = kk + c + j;
        p[j] = (gr < B && gc < K) ? (__bf16)X[(long)gr * K + gc]
                                  : (__bf16)0.f;
      }
      *reinterpret_cast<bf16x4*>(&xs[r * XS + c]) = p;
    }
    __syncthreads();
    const int kmax = min(BK, K - kk);

    auto load_wtile = [&](int t, float4* vr) {
#pragma unroll
      for (int i = 0; i < WCH; ++i) {
        const int ct = wave + NWAVES * t;
        const int col = ct * 16 + wc_row[i];
        const int k = wc_k[i];
        float4 v = {0.f, 0.f, 0.f, 0.f};
        if (ct < ntiles && col < N) {
          const float* wrow = W + (long)col * K + kk;
          if (k + 3 < kmax) {
            v = *reinterpret_cast<const float4*>(wrow + k);
          } else {
            if (k + 0 < kmax) v.x = wrow[k + 0];
            if (k + 1 < kmax) v.y = wrow[k + 1];
            if (k + 2 < kmax) v.z = wrow[k + 2];
            if (k + 3 < kmax) v.w = wrow[k + 3];
          }
        }
        vr[i] = v;
      }
    };
    auto write_wtile = [&](const float4* vr) {
#pragma unroll
      for (int i = 0; i < WCH; ++i) {
        const int o = wc_row[i] * XS + wc_k[i];
        // wc_k multiples of 4 and XS even => 8-byte aligned: one
        // ds_write_b64 instead of four 2-byte stores
        bf16x4 p = {(__bf16)vr[i].x, (__bf16)vr[i].y, (__bf16)vr[i].z,
                    (__bf16)vr[i].w};
        *reinterpret_cast<bf16x4*>(&wsw[o]) = p;
      }
    };

    float4 wa[WCH], wb[WCH];
    load_wtile(0, wa);
#pragma unroll
    for (int t = 0; t < NT_MAX; ++t) {
      const int ct = wave + NWAVES * t;
      if (ct >= ntiles) break;
      if (t % 2 == 0) {
        write_wtile(wa);
        if (t + 1 < NT_MAX) load_wtile(t + 1, wb);
      } else {
        write_wtile(wb);
        if (t + 1 < NT_MAX) load_wtile(t + 1, wa);
      }
      // two K=32 MFMAs per 64-wide tile; each lane's operand run is 8
      // contiguous bf16 = one b128 LDS read
      bf16x8 a0 = *reinterpret_cast<const bf16x8*>(&xs[l15 * XS + l4 * 8]);
      bf16x8 a1 =
          *reinterpret_cast<const bf16x8*>(&xs[l15 * XS + 32 + l4 * 8]);
      bf16x8 b0 = *reinterpret_cast<const bf16x8*>(&wsw[l15 * XS + l4 * 8]);
      bf16x8 b1 =
          *reinterpret_cast<const bf16x8*>(&wsw[l15 * XS + 32 + l4 * 8]);
      acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc[t],
                                                       0, 0, 0);
      acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acc[t],
                                                       0, 0, 0);
    }
    __syncthreads();
  }

  // ---- fp32 epilogue: bias + LN + act (identical to the f32 kernel) ----
  float zrow[NT_MAX][4];
  float psum[4] = {0.f, 0.f, 0.f, 0.f}, psq[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
  for (int t = 0; t < NT_MAX; ++t) {
    const int ct = wave + NWAVES * t;
    const int col = ct * 16 + l15;
    const bool colv = (ct < ntiles) && (col < N);
    const float bv = (colv && bias) ? bias[col] : 0.f;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float z = colv ? acc[t][r] + bv : 0.f;
      zrow[t][r] = z;
      psum[r] += z;
      psq[r] += z * z;
    }
  }

  if (with_ln) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      psum[r] = group16_sum(psum[r]);
      psq[r] = group16_sum(psq[r]);
    }
    if (l15 == 0) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        rowstat[(wave * 16 + l4 * 4 + r) * 2 + 0] = psum[r];
        rowstat[(wave * 16 + l4 * 4 + r) * 2 + 1] = psq[r];
      }
    }
    __syncthreads();
    if (tid < 16) {
      float sa = 0.f, q = 0.f;
#pragma unroll
      for (int w = 0; w < NWAVES; ++w) {
        sa += rowstat[(w * 16 + tid) * 2 + 0];
        q += rowstat[(w * 16 + tid) * 2 + 1];
      }
      const float mean = sa / N;
      float var = q / N - mean * mean;
      const float rstd = rsqrtf(fmaxf(var, 0.f) + 1e-5f);
      rowmv[tid * 2 + 0] = mean;
      rowmv[tid * 2 + 1] = rstd;
      const int grow = row0 + tid;
      if (grow < B && RSTD) RSTD[grow] = rstd;
    }
    __syncthreads();
  }

#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int lrow = l4 * 4 + r;
    const int grow = row0 + lrow;
    if (grow >= B) continue;
    float mean = 0.f, rstd = 1.f;
    if (with_ln) {
      mean = rowmv[lrow * 2 + 0];
      rstd = rowmv[lrow * 2 + 1];
    }
#pragma unroll
    for (int t = 0; t < NT_MAX; ++t) {
      const int ct = wave + NWAVES * t;
      const int col = ct * 16 + l15;
      if (ct < ntiles && col < N) {
        float z = zrow[t][r];
        float out;
        if (with_ln) {
          const float zh = (z - mean) * rstd;
          ZHAT[(long)grow * N + col] = zh;
          out = apply_act(gamma[col] * zh + beta[col], act);
        } else {
          out = apply_act(z, act);
        }
        Y[(long)grow * N + col] = out;
      }
    }
  }
}

// C (M,N) = A (M,K) @ B (K,N), fp32 interfaces, bf16 MFMA (dX = dZ @ W).
extern "C" __global__ __launch_bounds__(256) void gemm_bf16_nn_kernel(
    const float* __restrict__ A, const float* __restrict__ B,
    float* __restrict__ C, int M, int K, int N) {
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;
  const int row0 = blockIdx.x * 16;
  const int col0 = (blockIdx.y * 4 + wave) * 16;

  __shared__ __bf16 as[2][16 * (BK + XPAD)];
  const int XS = BK + XPAD;
  const int col = col0 + l15;
  const int st_r = (tid * 4) >> 6, st_c = (tid * 4) & 63;
  auto load_a = [&](int kk, float4* v) {
    const int gr = row0 + st_r;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int gc = kk + st_c + j;
      (&v->x)[j] = (gr < M && gc < K) ? A[(long)gr * K + gc] : 0.f;
    }
  };
  auto store_a = [&](int buf, const float4* v) {
    // st_c is a multiple of 4 and XS is even: one packed 8-byte store
    bf16x4 p = {(__bf16)v->x, (__bf16)v->y, (__bf16)v->z, (__bf16)v->w};
    *reinterpret_cast<bf16x4*>(&as[buf][st_r * XS + st_c]) = p;
  };
  f32x4 acc = (f32x4){0.f, 0.f, 0.f, 0.f};
  float4 areg;
  load_a(0, &areg);
  store_a(0, &areg);
  __syncthreads();
  int cur = 0;
  for (int kk = 0; kk < K; kk += BK) {
    const int kmax = min(BK, K - kk);
    if (kk + BK < K) load_a(kk + BK, &areg);
    // front-load the 16 strided B scalars (k-run of 8 per half), cast to
    // bf16 fragments
    bf16x8 b0, b1;
#pragma unroll
    for (int r = 0; r < 8; ++r) {
      const int k0 = l4 * 8 + r;
      const int k1 = 32 + l4 * 8 + r;
      b0[r] = (col < N && k0 < kmax) ? (__bf16)B[(long)(kk + k0) * N + col]
                                     : (__bf16)0.f;
      b1[r] = (col < N && k1 < kmax) ? (__bf16)B[(long)(kk + k1) * N + col]
                                     : (__bf16)0.f;
    }
    bf16x8 a0 = *reinterpret_cast<const bf16x8*>(&as[cur][l15 * XS + l4 * 8]);
    bf16x8 a1 =
        *reinterpret_cast<const bf16x8*>(&as[cur][l15 * XS + 32 + l4 * 8]);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc, 0, 0, 0);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acc, 0, 0, 0);
    if (kk + BK < K) {
      store_a(1 - cur, &areg);
      cur = 1 - cur;
    }
    __syncthreads();
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int orow = row0 + l4 * 4 + r;
    const int ocol = col0 + l15;
    if (orow < M && ocol < N) C[(long)orow * N + ocol] = acc[r];
  }
}

// C (M,N) += A^T @ B with A (K,M), B (K,N) fp32 row-major, bf16 MFMA
// (dW = dZ^T @ X accumulated straight into the flat fp32 grad pool).
extern "C" __global__ __launch_bounds__(256) void gemm_bf16_tn_kernel(
    const float* __restrict__ A, const float* __restrict__ B,
    float* __restrict__ C, int M, int K, int N, int accumulate) {
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;
  const int row0 = blockIdx.x * 16;
  const int col0 = (blockIdx.y * 4 + wave) * 16;
  if (col0 >= N) return;

  const int arow = row0 + l15;
  const int bcol = col0 + l15;
  f32x4 acc = (f32x4){0.f, 0.f, 0.f, 0.f};
  for (int kk = 0; kk < K; kk += 32) {
    bf16x8 af, bf;
#pragma unroll
    for (int r = 0; r < 8; ++r) {
      const int k = kk + l4 * 8 + r;
      af[r] = (arow < M && k < K) ? (__bf16)A[(long)k * M + arow]
                                  : (__bf16)0.f;
      bf[r] = (bcol < N && k < K) ? (__bf16)B[(long)k * N + bcol]
                                  : (__bf16)0.f;
    }
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc, 0, 0, 0);
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int orow = row0 + l4 * 4 + r;
    const int ocol = col0 + l15;
    if (orow < M && ocol < N) {
      const long idx = (long)orow * N + ocol;
      C[idx] = accumulate ? C[idx] + acc[r] : acc[r];
    }
  }
}

// ---------------------------------------------------------------------------
// Whole-chain MLP forward, bf16 compute (the bf16-mode counterpart of
// mlp_chain_fwd_kernel in fused_linear.hip): activations ping-pong in
// fp32 LDS (cast to bf16 at fragment assembly — identical numerics to
// the per-layer bf16 path, which casts on staging), W subtiles staged
// as packed bf16, GEMMs on v_mfma_f32_16x16x32_bf16, fp32 LN epilogue.
// ---------------------------------------------------------------------------

#define CHAIN_MAX 4
#define CNW 16          // waves per block (1024 threads)
#define CNT 3           // max 16-col tiles per wave (16*16*3 = 768 >= 512)
#define XMAX 576
#define XP (XMAX + 1)
#define WXS (BK + XPAD)  // bf16 W row stride

struct ChainArgsB {
  const float* W[CHAIN_MAX];
  const float* bias[CHAIN_MAX];
  const float* gamma[CHAIN_MAX];
  const float* beta[CHAIN_MAX];
  float* Y[CHAIN_MAX];
  float* ZHAT[CHAIN_MAX];
  float* RSTD[CHAIN_MAX];
  int dims[CHAIN_MAX + 1];
  int act[CHAIN_MAX];
  int with_ln[CHAIN_MAX];
  int L;
  int B;
};

extern "C" __global__ __launch_bounds__(1024) void mlp_chain_bf16_fwd_kernel(
    const float* __restrict__ X, ChainArgsB args) {
  extern __shared__ float smem[];
  float* buf0 = smem;                       // [16][XP] fp32 activations
  float* buf1 = smem + 16 * XP;
  __bf16* ws = reinterpret_cast<__bf16*>(smem + 2 * 16 * XP);
  float* rowstat = smem + 2 * 16 * XP
      + (CNW * 16 * WXS + 1) / 2;           // after bf16 region (in floats)
  float* rowmv = rowstat + CNW * 16 * 2;

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;
  const int row0 = blockIdx.x * 16;
  const int B = args.B;

  const int wc_row[WCH] = {(lane + 0 * WAVE) / (BK / 4),
                           (lane + 1 * WAVE) / (BK / 4),
                           (lane + 2 * WAVE) / (BK / 4),
                           (lane + 3 * WAVE) / (BK / 4)};
  const int wc_k[WCH] = {((lane + 0 * WAVE) % (BK / 4)) * 4,
                         ((lane + 1 * WAVE) % (BK / 4)) * 4,
                         ((lane + 2 * WAVE) % (BK / 4)) * 4,
                         ((lane + 3 * WAVE) % (BK / 4)) * 4};
  __bf16* wsw = ws + wave * 16 * WXS;

  // stage the input (zero-padded to a BK boundary), fp32
  {
    const int K0 = args.dims[0];
    const int Kpad = ((K0 + BK - 1) / BK) * BK;
    for (int idx = tid; idx < 16 * Kpad; idx += 1024) {
      const int r = idx / Kpad, c = idx % Kpad;
      const int gr = row0 + r;
      buf0[r * XP + c] = (gr < B && c < K0) ? X[(long)gr * K0 + c] : 0.f;
    }
  }
  __syncthreads();

  float* xb = buf0;
  float* yb = buf1;

  for (int l = 0; l < args.L; ++l) {
    const int K = args.dims[l];
    const int N = args.dims[l + 1];
    const int ntiles = (N + 15) >> 4;
    const float* W = args.W[l];
    const int with_ln = args.with_ln[l];

    f32x4 acc[CNT];
#pragma unroll
    for (int t = 0; t < CNT; ++t) acc[t] = (f32x4){0.f, 0.f, 0.f, 0.f};

    for (int kk = 0; kk < K; kk += BK) {
      const int kmax = min(BK, K - kk);
      auto load_wtile = [&](int t, float4* vr) {
#pragma unroll
        for (int i = 0; i < WCH; ++i) {
          const int ct = wave + CNW * t;
          const int col = ct * 16 + wc_row[i];
          const int k = wc_k[i];
          float4 v = {0.f, 0.f, 0.f, 0.f};
          if (ct < ntiles && col < N) {
            const float* wrow = W + (long)col * K + kk;
            if (k + 3 < kmax) {
              v = *reinterpret_cast<const float4*>(wrow + k);
            } else {
              if (k + 0 < kmax) v.x = wrow[k + 0];
              if (k + 1 < kmax) v.y = wrow[k + 1];
              if (k + 2 < kmax) v.z = wrow[k + 2];
              if (k + 3 < kmax) v.w = wrow[k + 3];
            }
          }
          vr[i] = v;
        }
      };
      auto write_wtile = [&](const float4* vr) {
#pragma unroll
        for (int i = 0; i < WCH; ++i) {
          bf16x4 p = {(__bf16)vr[i].x, (__bf16)vr[i].y, (__bf16)vr[i].z,
                      (__bf16)vr[i].w};
          *reinterpret_cast<bf16x4*>(&wsw[wc_row[i] * WXS + wc_k[i]]) = p;
        }
      };
      float4 wa[WCH], wb[WCH];
      load_wtile(0, wa);
#pragma unroll
      for (int t = 0; t < CNT; ++t) {
        const int ct = wave + CNW * t;
        if (ct >= ntiles) break;
        if (t % 2 == 0) {
          write_wtile(wa);
          if (t + 1 < CNT) load_wtile(t + 1, wb);
        } else {
          write_wtile(wb);
          if (t + 1 < CNT) load_wtile(t + 1, wa);
        }
        // A fragments: fp32 activations -> bf16 at assembly
        bf16x8 a0, a1, b0, b1;
#pragma unroll
        for (int r = 0; r < 8; ++r) {
          a0[r] = (__bf16)xb[l15 * XP + kk + l4 * 8 + r];
          a1[r] = (__bf16)xb[l15 * XP + kk + 32 + l4 * 8 + r];
        }
        b0 = *reinterpret_cast<const bf16x8*>(&wsw[l15 * WXS + l4 * 8]);
        b1 = *reinterpret_cast<const bf16x8*>(
            &wsw[l15 * WXS + 32 + l4 * 8]);
        acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc[t],
                                                         0, 0, 0);
        acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acc[t],
                                                         0, 0, 0);
      }
    }

    // ---- fp32 epilogue: bias + LN + act; write HBM + the LDS ping ----
    const float* bias = args.bias[l];
    const float* gamma = args.gamma[l];
    const float* beta = args.beta[l];
    float zrow[CNT][4];
    float psum[4] = {0.f, 0.f, 0.f, 0.f}, psq[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int t = 0; t < CNT; ++t) {
      const int ct = wave + CNW * t;
      const int col = ct * 16 + l15;
      const bool colv = (ct < ntiles) && (col < N);
      const float bv = (colv && bias) ? bias[col] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float z = colv ? acc[t][r] + bv : 0.f;
        zrow[t][r] = z;
        psum[r] += z;
        psq[r] += z * z;
      }
    }
    if (with_ln) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        psum[r] = group16_sum(psum[r]);
        psq[r] = group16_sum(psq[r]);
      }
      if (l15 == 0) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          rowstat[(wave * 16 + l4 * 4 + r) * 2 + 0] = psum[r];
          rowstat[(wave * 16 + l4 * 4 + r) * 2 + 1] = psq[r];
        }
      }
      __syncthreads();
      if (tid < 16) {
        float sa = 0.f, q = 0.f;
#pragma unroll
        for (int w = 0; w < CNW; ++w) {
          sa += rowstat[(w * 16 + tid) * 2 + 0];
          q += rowstat[(w * 16 + tid) * 2 + 1];
        }
        const float mean = sa / N;
        float var = q / N - mean * mean;
        const float rstd = rsqrtf(fmaxf(var, 0.f) + 1e-5f);
        rowmv[tid * 2 + 0] = mean;
        rowmv[tid * 2 + 1] = rstd;
        const int grow = row0 + tid;
        if (grow < B && args.RSTD[l]) args.RSTD[l][grow] = rstd;
      }
      __syncthreads();
    }
    {
      const int Npad = ((N + BK - 1) / BK) * BK;
      const int padw = Npad - N;
      if (padw > 0) {
        for (int idx = tid; idx < 16 * padw; idx += 1024) {
          const int r = idx / padw, c = N + idx % padw;
          yb[r * XP + c] = 0.f;
        }
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int lrow = l4 * 4 + r;
      const int grow = row0 + lrow;
      float mean = 0.f, rstd = 1.f;
      if (with_ln) {
        mean = rowmv[lrow * 2 + 0];
        rstd = rowmv[lrow * 2 + 1];
      }
#pragma unroll
      for (int t = 0; t < CNT; ++t) {
        const int ct = wave + CNW * t;
        const int col = ct * 16 + l15;
        if (ct < ntiles && col < N) {
          float z = zrow[t][r];
          float out;
          if (with_ln) {
            const float zh = (z - mean) * rstd;
            if (grow < B) args.ZHAT[l][(long)grow * N + col] = zh;
            out = apply_act(gamma[col] * zh + beta[col], args.act[l]);
          } else {
            out = apply_act(z, args.act[l]);
          }
          yb[lrow * XP + col] = out;
          if (grow < B) args.Y[l][(long)grow * N + col] = out;
        }
      }
    }
    __syncthreads();
    float* tmp = xb; xb = yb; yb = tmp;
  }
}
