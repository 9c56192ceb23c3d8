// Fused Linear(+LayerNorm)(+activation) forward and backward — the RL
// actor/critic MLP hot op (reference composes
// Linear -> LayerNorm -> ELU as separate torch ops, enet_sac.py:436-444).
//
// Forward kernel: one workgroup (8 waves, 512 threads) owns a 16-row slab of
// the batch and ALL output columns (so the LayerNorm row statistics never
// leave the workgroup). The GEMM runs on v_mfma_f32_16x16x4_f32 (exact f32,
// §3 of the CDNA4 guide; no TF32 on gfx950) with the X tile staged in LDS
// (+1 pad, conflict free) and each wave register-prefetching its NEXT
// 16-col W subtile (coalesced float4 bursts) while the current one runs on
// the matrix cores — the T14 issue-early/write-late split of the CDNA4
// guide, sized for RL shapes where W is L2-resident and the kernel is
// latency-bound. The epilogue fuses bias + LayerNorm (shfl_xor 16-lane row
// reductions + one LDS cross-wave combine) + activation, storing y, zhat,
// rstd for backward. No separate normalization kernel, no extra HBM round
// trip.
//
// Backward: (1) ln_act_bwd_kernel — act'(y) + LayerNorm backward (row
// reductions in-workgroup) + dgamma/dbeta column partials; (2)/(3) are MFMA
// GEMMs in gemm_f32.hip.

#include "common.h"

#define NWAVES 8
#define NT_MAX 12          // max 16-col tiles per wave => N <= 16*NWAVES*NT_MAX = 1536
#define BK 64              // K-tile staged in LDS
#define WCH (16 * (BK / 4) / WAVE)  // float4 chunks per lane per W subtile (=4)

extern "C" __global__ __launch_bounds__(512) void fused_linear_fwd_kernel(
    const float* __restrict__ X,    // (B, K)
    const float* __restrict__ W,    // (N, K) row-major
    const float* __restrict__ bias, // (N) or null
    const float* __restrict__ gamma,// (N) or null (=> no LN)
    const float* __restrict__ beta, // (N)
    float* __restrict__ Y,          // (B, N)
    float* __restrict__ ZHAT,       // (B, N) (LN only)
    float* __restrict__ RSTD,       // (B,)   (LN only)
    int B, int K, int N, int act, int with_ln) {
  // one __shared__ object (CDNA4 guide §5.5 item 4a): X tile + NWAVES
  // per-wave W subtiles (stride BK+1: conflict-free b32 fragment reads)
  __shared__ float smem[16 * (BK + 1) + NWAVES * 16 * (BK + 1)
                        + NWAVES * 16 * 2 + 16 * 2];
  float* xs = smem;                               // [16][BK+1]
  float* ws = smem + 16 * (BK + 1);               // [wave][16][BK+1]
  float* rowstat = ws + NWAVES * 16 * (BK + 1);   // [NWAVES][16 rows][2]
  float* rowmv = rowstat + NWAVES * 16 * 2;       // [16][2] mean, rstd

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;          // 0..3
  const int row0 = blockIdx.x * 16;  // first batch row of this block

  const int ntiles = (N + 15) >> 4;  // 16-col tiles over N

  // per-lane W-chunk coordinates (fixed across tiles)
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

  float* wsw = ws + wave * 16 * (BK + 1);

  for (int kk = 0; kk < K; kk += BK) {
    // stage X[row0..row0+15][kk..kk+BK) into LDS (zero-padded)
    for (int idx = tid; idx < 16 * BK; idx += 512) {
      int r = idx / BK, c = idx % BK;
      int gr = row0 + r, gc = kk + c;
      xs[r * (BK + 1) + c] =
          (gr < B && gc < K) ? X[(long)gr * K + gc] : 0.f;
    }
    __syncthreads();
    const int kmax = min(BK, K - kk);

    // coalesced, guarded float4 load of one W subtile into registers
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
        const int o = wc_row[i] * (BK + 1) + wc_k[i];
        wsw[o + 0] = vr[i].x;
        wsw[o + 1] = vr[i].y;
        wsw[o + 2] = vr[i].z;
        wsw[o + 3] = vr[i].w;
      }
    };

    // software pipeline over this wave's column tiles: write tile t to
    // LDS, issue tile t+1's global loads, run tile t's MFMAs (the global
    // latency hides under the MFMA + ds traffic — T14)
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
      // front-load ALL LDS operand reads, then run the MFMA chain with a
      // split accumulator: interleaved read/MFMA makes the compiler emit
      // an lgkmcnt(0) wait per MFMA pair, exposing LDS latency each time
      float ar[BK / 4], br[BK / 4];
#pragma unroll
      for (int k4 = 0; k4 < BK / 4; ++k4) {
        const int k = k4 * 4 + l4;
        ar[k4] = xs[l15 * (BK + 1) + k];
        br[k4] = wsw[l15 * (BK + 1) + k];
      }
      // MFMA is a whole-wave op: never predicate it per-lane; edge
      // tiles contribute zeros through the zero-padded LDS operands.
      f32x4 acc2 = (f32x4){0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int k4 = 0; k4 < BK / 4; k4 += 2) {
        acc[t] = __builtin_amdgcn_mfma_f32_16x16x4f32(ar[k4], br[k4],
                                                      acc[t], 0, 0, 0);
        acc2 = __builtin_amdgcn_mfma_f32_16x16x4f32(ar[k4 + 1], br[k4 + 1],
                                                    acc2, 0, 0, 0);
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) acc[t][r] += acc2[r];
    }
    __syncthreads();
  }

  // ---- epilogue: bias + LN + act ----
  // acc[t][r] holds z[row = row0 + l4*4 + r][col = (wave+NWAVES*t)*16 + l15]
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
    // 16-lane reduction: lanes sharing l4 cover 16 distinct cols
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

// LayerNorm+activation backward: dz from dy, plus dgamma/dbeta column sums.
// One workgroup per batch row (block reductions over the N columns).
extern "C" __global__ __launch_bounds__(256) void ln_act_bwd_kernel(
    const float* __restrict__ DY,   // (B, N)
    const float* __restrict__ Yv,   // (B, N) activation output
    const float* __restrict__ ZHAT, // (B, N)
    const float* __restrict__ RSTD, // (B,)
    const float* __restrict__ gamma,// (N)
    float* __restrict__ DZ,         // (B, N)
    float* __restrict__ DGAMMA,     // (N) pre-zeroed, atomically accumulated
    float* __restrict__ DBETA,      // (N)
    int B, int N, int act, int with_ln) {
  __shared__ float red[2 * 4];  // per-wave partial sums (m1, m2)
  const int row = blockIdx.x;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;

  if (!with_ln) {
    for (int c = tid; c < N; c += blockDim.x) {
      const long i = (long)row * N + c;
      DZ[i] = DY[i] * act_grad_from_y(Yv[i], act);
    }
    return;
  }

  float m1 = 0.f, m2 = 0.f;
  for (int c = tid; c < N; c += blockDim.x) {
    const long i = (long)row * N + c;
    const float dh = DY[i] * act_grad_from_y(Yv[i], act);
    const float zh = ZHAT[i];
    const float dzh = dh * gamma[c];
    m1 += dzh;
    m2 += dzh * zh;
    atomicAdd(&DGAMMA[c], dh * zh);
    atomicAdd(&DBETA[c], dh);
  }
  m1 = wave_sum(m1);
  m2 = wave_sum(m2);
  if ((tid & 63) == 0) {
    red[wave * 2 + 0] = m1;
    red[wave * 2 + 1] = m2;
  }
  __syncthreads();
  float M1 = 0.f, M2 = 0.f;
  const int nw = (blockDim.x + 63) / 64;
  for (int w = 0; w < nw; ++w) {
    M1 += red[w * 2 + 0];
    M2 += red[w * 2 + 1];
  }
  M1 /= N;
  M2 /= N;
  const float rstd = RSTD[row];
  for (int c = tid; c < N; c += blockDim.x) {
    const long i = (long)row * N + c;
    const float dh = DY[i] * act_grad_from_y(Yv[i], act);
    const float dzh = dh * gamma[c];
    DZ[i] = rstd * (dzh - M1 - ZHAT[i] * M2);
  }
}

// ---------------------------------------------------------------------------
// Whole-chain MLP forward: up to CHAIN_MAX Linear(+LN)(+act) layers in ONE
// kernel. Activations ping-pong between two LDS buffers (no HBM round trip
// and no kernel launch between layers; the per-k-tile X staging barriers of
// the single-layer kernel disappear — X is staged once per layer). Each
// layer still stores y / zhat / rstd to HBM for the (per-layer) backward.
// Dynamic LDS (~108 KB) => one block per CU, which is fine: the batch-16
// row slabs give only ceil(B/16) blocks and the kernel is latency-bound.
// ---------------------------------------------------------------------------

#define CHAIN_MAX 4
#define CNW 16          // waves per block (1024 threads): halves the per-wave subtile chain
#define CNT 3           // max 16-col tiles per wave (16*16*3 = 768 >= 512)
#define XMAX 576              // max staged width (512 + one BK of zero pad)
#define XP (XMAX + 1)

struct ChainArgs {
  const float* W[CHAIN_MAX];
  const float* bias[CHAIN_MAX];
  const float* gamma[CHAIN_MAX];
  const float* beta[CHAIN_MAX];
  float* Y[CHAIN_MAX];
  float* ZHAT[CHAIN_MAX];
  float* RSTD[CHAIN_MAX];
  int dims[CHAIN_MAX + 1];    // dims[0]=K0 input, dims[l+1]=N_l
  int act[CHAIN_MAX];
  int with_ln[CHAIN_MAX];
  int L;
  int B;
};

extern "C" __global__ __launch_bounds__(1024) void mlp_chain_fwd_kernel(
    const float* __restrict__ X, ChainArgs args) {
  extern __shared__ float smem[];
  float* buf0 = smem;                       // [16][XP]
  float* buf1 = smem + 16 * XP;             // [16][XP]
  float* ws = smem + 2 * 16 * XP;           // [CNW][16][BK+1]
  float* rowstat = ws + CNW * 16 * (BK + 1);
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
  float* wsw = ws + wave * 16 * (BK + 1);

  // stage the input (zero-padded to a BK boundary)
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
          const int o = wc_row[i] * (BK + 1) + wc_k[i];
          wsw[o + 0] = vr[i].x;
          wsw[o + 1] = vr[i].y;
          wsw[o + 2] = vr[i].z;
          wsw[o + 3] = vr[i].w;
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
        // front-load all LDS operand reads (see single-layer kernel note)
        float ar[BK / 4], br[BK / 4];
#pragma unroll
        for (int k4 = 0; k4 < BK / 4; ++k4) {
          ar[k4] = xb[l15 * XP + kk + k4 * 4 + l4];
          br[k4] = wsw[l15 * (BK + 1) + k4 * 4 + l4];
        }
        f32x4 acc2 = (f32x4){0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int k4 = 0; k4 < BK / 4; k4 += 2) {
          acc[t] = __builtin_amdgcn_mfma_f32_16x16x4f32(ar[k4], br[k4],
                                                        acc[t], 0, 0, 0);
          acc2 = __builtin_amdgcn_mfma_f32_16x16x4f32(ar[k4 + 1],
                                                      br[k4 + 1], acc2,
                                                      0, 0, 0);
        }
#pragma unroll
        for (int r = 0; r < 4; ++r) acc[t][r] += acc2[r];
      }
    }

    // ---- epilogue: bias + LN + act; write HBM + the LDS ping buffer ----
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
    // zero the pad region of the ping buffer for the next layer's reads
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
