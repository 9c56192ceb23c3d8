// General L-BFGS two-loop recursion — N6/N7 of SURVEY.md §2.2.
//
// Reference spec: `lbfgsnew.py:637-651` (search-direction two-loop) and
// `autograd_tools.inv_hessian_mult[_mat]` (:35-66, the influence-map
// iHVP). The host composition does 4h dot/axpy launches PLUS a host
// sync per dot (`float(...)`); this kernel runs the WHOLE recursion for
// m right-hand sides in ONE launch: one 256-thread workgroup per
// column, grid-stride dot/axpy passes over the n parameters, block
// reductions through wave DPP sums + LDS.
//
// Layouts: Y, S (h, n) newest-last; Q, R (m, n) ROW-major (each row one
// RHS, coalesced along n); ro (h) precomputed 1/(y_i·s_i) host-side
// (lets callers zero-out filtered pairs); gamma = the H_diag scale.

#include "common.h"

#define TL_HMAX 16

extern "C" __global__ __launch_bounds__(256) void two_loop_kernel(
    const float* __restrict__ Y, const float* __restrict__ S,
    const float* __restrict__ Q, float* __restrict__ R,
    const float* __restrict__ ro, float gamma, int h, long n, int m) {
  __shared__ float red[4];
  __shared__ float al[TL_HMAX];
  __shared__ float cur;
  const int c = blockIdx.x;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  float* q = R + (long)c * n;
  const float* qin = Q + (long)c * n;
  for (long j = tid; j < n; j += blockDim.x) q[j] = qin[j];
  __syncthreads();

  auto block_dot = [&](const float* a, const float* b) {
    float s = 0.f;
    for (long j = tid; j < n; j += blockDim.x) s += a[j] * b[j];
    s = wave_sum(s);
    if ((tid & 63) == 0) red[wave] = s;
    __syncthreads();
    const float tot = red[0] + red[1] + red[2] + red[3];
    __syncthreads();
    return tot;
  };

  for (int i = h - 1; i >= 0; --i) {
    const float d = block_dot(&S[(long)i * n], q);
    if (tid == 0) {
      al[i] = d * ro[i];
      cur = al[i];
    }
    __syncthreads();
    const float a = cur;
    for (long j = tid; j < n; j += blockDim.x)
      q[j] -= a * Y[(long)i * n + j];
    __syncthreads();
  }
  for (long j = tid; j < n; j += blockDim.x) q[j] *= gamma;
  __syncthreads();
  for (int i = 0; i < h; ++i) {
    const float d = block_dot(&Y[(long)i * n], q);
    if (tid == 0) cur = al[i] - d * ro[i];
    __syncthreads();
    const float a = cur;
    for (long j = tid; j < n; j += blockDim.x)
      q[j] += a * S[(long)i * n + j];
    __syncthreads();
  }
}
