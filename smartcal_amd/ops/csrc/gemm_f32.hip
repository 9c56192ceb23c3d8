// fp32 MFMA GEMMs for the fused-linear backward pass (dX, dW) and other
// small fp32 matmuls. v_mfma_f32_16x16x4_f32 — exact f32 (bitwise an fmaf
// chain), 16x16 tile per wave, 4-deep K per instruction.
//
// These shapes are RL-sized (M,N <= ~5k, K <= ~5k but typically B=64..256):
// operands are L2-resident, so operand fragments are read straight from
// global (coalesced on the 16-consecutive-column axis); the fused forward
// kernel in fused_linear.hip is the LDS-staged showpiece.

#include "common.h"

// C (M,N) = A (M,K) @ B (K,N), all row-major.
extern "C" __global__ __launch_bounds__(256) void gemm_f32_nn_kernel(
    const float* __restrict__ A, const float* __restrict__ B,
    float* __restrict__ C, int M, int K, int N) {
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;
  const int row0 = blockIdx.x * 16;
  const int col0 = (blockIdx.y * 4 + wave) * 16;
  // NOTE: no early return — all waves must reach the barriers below;
  // out-of-range waves compute on zeros and skip their stores.

  // A tiles double-buffered in LDS (A fragment reads are row-scattered in
  // global): the global loads of chunk i+1 are issued into registers
  // before chunk i's MFMAs, so their latency hides under the matrix-core
  // work — one barrier per chunk instead of two. B fragment reads are
  // naturally coalesced (consecutive lanes -> consecutive columns), so B
  // streams from L2.
  __shared__ float as[2][16 * 68];  // [16][64+4] pad 4: conflict-free reads
  const int col = col0 + l15;   // B fragment col
  // each thread stages 4 A elements per chunk (16*64/256)
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
#pragma unroll
    for (int j = 0; j < 4; ++j)
      as[buf][st_r * 68 + st_c + j] = (&v->x)[j];
  };
  f32x4 acc = (f32x4){0.f, 0.f, 0.f, 0.f};
  float4 areg;
  load_a(0, &areg);
  store_a(0, &areg);
  __syncthreads();
  int cur = 0;
  for (int kk = 0; kk < K; kk += 64) {
    const int kmax = min(64, K - kk);
    if (kk + 64 < K) load_a(kk + 64, &areg);  // prefetch next chunk
    // issue ALL 16 B loads before the MFMA chain: interleaving them
    // makes the compiler emit waitcnt vmcnt(0) before every MFMA, which
    // exposes a full L2 round trip per instruction (~24 us/kernel);
    // front-loading brings the chunk down to one latency + the MFMAs
    float breg[16];
#pragma unroll
    for (int k4 = 0; k4 < 16; ++k4) {
      const int k = k4 * 4 + l4;
      breg[k4] = (col < N && k < kmax) ? B[(long)(kk + k) * N + col] : 0.f;
    }
#pragma unroll
    for (int k4 = 0; k4 < 16; ++k4) {
      const float a = as[cur][l15 * 68 + k4 * 4 + l4];
      acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, breg[k4], acc, 0, 0, 0);
    }
    if (kk + 64 < K) {
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

// C (M,N) = A^T @ B with A (K,M), B (K,N) row-major (e.g. dW = dZ^T @ X).
// accumulate: C += result (direct-accumulate into a pre-zeroed flat grad
// pool — removes the separate autograd add/zero kernels per parameter).
extern "C" __global__ __launch_bounds__(256) void gemm_f32_tn_kernel(
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

  const int arow = row0 + l15;  // output row = column of A
  const int bcol = col0 + l15;
  f32x4 acc = (f32x4){0.f, 0.f, 0.f, 0.f};
  // front-load operands in chunks of 8 k-steps so the MFMAs don't stall
  // on a vmcnt(0) per instruction (see gemm_nn note)
  for (int kk = 0; kk < K; kk += 32) {
    float areg[8], breg[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int k = kk + j * 4 + l4;
      areg[j] = (arow < M && k < K) ? A[(long)k * M + arow] : 0.f;
      breg[j] = (bcol < N && k < K) ? B[(long)k * N + bcol] : 0.f;
    }
#pragma unroll
    for (int j = 0; j < 8; ++j)
      acc = __builtin_amdgcn_mfma_f32_16x16x4f32(areg[j], breg[j], acc,
                                                 0, 0, 0);
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

// db (N) = column sums of DZ (B, N). Block = 256 threads covering 64
// columns x 4 row-stripes (coalesced: consecutive threads -> consecutive
// columns), combined through LDS. accumulate: OUT += sums.
extern "C" __global__ __launch_bounds__(256) void colsum_kernel(
    const float* __restrict__ DZ, float* __restrict__ OUT, int B, int N,
    int accumulate) {
  __shared__ float red[4 * 64];
  const int c = blockIdx.x * 64 + (threadIdx.x & 63);
  const int stripe = threadIdx.x >> 6;
  float s = 0.f;
  if (c < N)
    for (int r = stripe; r < B; r += 4) s += DZ[(long)r * N + c];
  red[stripe * 64 + (threadIdx.x & 63)] = s;
  __syncthreads();
  if (threadIdx.x < 64 && c < N) {
    const float s4 = red[threadIdx.x] + red[64 + threadIdx.x] +
                     red[128 + threadIdx.x] + red[192 + threadIdx.x];
    OUT[c] = accumulate ? OUT[c] + s4 : s4;
  }
}
