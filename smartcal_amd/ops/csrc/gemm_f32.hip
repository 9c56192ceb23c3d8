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
  if (col0 >= N) return;

  const int row = row0 + l15;   // A fragment row
  const int col = col0 + l15;   // B fragment col
  f32x4 acc = (f32x4){0.f, 0.f, 0.f, 0.f};
  for (int kk = 0; kk < K; kk += 4) {
    const int k = kk + l4;
    const float a = (row < M && k < K) ? A[(long)row * K + k] : 0.f;
    const float b = (col < N && k < K) ? B[(long)k * N + col] : 0.f;
    acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
  }
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int orow = row0 + l4 * 4 + r;
    const int ocol = col0 + l15;
    if (orow < M && ocol < N) C[(long)orow * N + ocol] = acc[r];
  }
}

// C (M,N) = A^T @ B with A (K,M), B (K,N) row-major (e.g. dW = dZ^T @ X).
extern "C" __global__ __launch_bounds__(256) void gemm_f32_tn_kernel(
    const float* __restrict__ A, const float* __restrict__ B,
    float* __restrict__ C, int M, int K, int N) {
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
    if (orow < M && ocol < N) C[(long)orow * N + ocol] = acc[r];
  }
}

// db (N) = column sums of DZ (B, N).
extern "C" __global__ void colsum_kernel(const float* __restrict__ DZ,
                                         float* __restrict__ OUT, int B,
                                         int N) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= N) return;
  float s = 0.f;
  for (int r = 0; r < B; ++r) s += DZ[(long)r * N + c];
  OUT[c] = s;
}
