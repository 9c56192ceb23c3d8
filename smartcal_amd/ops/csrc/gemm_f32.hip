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

  // A tile staged in LDS (A fragment reads are row-scattered in global);
  // B fragment reads are naturally coalesced (consecutive lanes ->
  // consecutive columns), so B streams from L2.
  __shared__ float as[16 * 68];  // [16][64+4] pad 4: conflict-free b32 reads
  const int col = col0 + l15;   // B fragment col
  f32x4 acc = (f32x4){0.f, 0.f, 0.f, 0.f};
  for (int kk = 0; kk < K; kk += 64) {
    const int kmax = min(64, K - kk);
    for (int idx = tid; idx < 16 * 64; idx += 256) {
      const int r = idx >> 6, c = idx & 63;
      const int gr = row0 + r, gc = kk + c;
      as[r * 68 + c] = (gr < M && gc < K) ? A[(long)gr * K + gc] : 0.f;
    }
    __syncthreads();
#pragma unroll
    for (int k4 = 0; k4 < 16; ++k4) {
      const int k = k4 * 4 + l4;
      const float a = as[l15 * 68 + k];
      const float b = (col < N && k < kmax) ? B[(long)(kk + k) * N + col]
                                            : 0.f;
      acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
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
  for (int kk = 0; kk < K; kk += 4) {
    const int k = kk + l4;
    const float a = (arow < M && k < K) ? A[(long)k * M + arow] : 0.f;
    const float b = (bcol < N && k < K) ? B[(long)k * N + bcol] : 0.f;
    acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
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
