// Calibration Hessian assembly (N9) — ONE launch.
//
// Reference spec: `Hessianres[_torch]` (calibration_tools.py:589-676), a
// K x (T·B) python loop of 2x2/4x4 kron accumulations into K x 4N x 4N.
// The torch rewrite (radio/hessian.hessianres) is ~10 batched
// einsum/index_add launches; this kernel fuses the whole assembly:
// grid (baseline-tiles x K), 16 threads per baseline each owning one
// element of the 4x4 blocks, t-loop in registers.
//
// Math (see hessianres docstring): per (k, baseline b=(p,q), slot t)
//   off  (p,q): [(i,a),(j,c)] += -conj(Ci)[i][j] * Res[a][c]
//   mirror (q,p): conj-transpose of the off block
//   diag (p,p): [(i,a),(j,a)] += (Ci G_q Ci^H)[j][i],  G_q = Jq^H Jq
//   diag (q,q): [(i,a),(j,a)] += (Ci^H G_p Ci)[j][i],  G_p = Jp^H Jp
// with Ci the column-major 2x2 view of C[k, t*B+b, :] and G_p/G_q
// t-independent (J is per solution interval). All scaled by 1/(B·T).
// Off/mirror blocks are unique per (p,q) -> direct stores; diagonal
// blocks accumulate across baselines -> float atomics.

#include "common.h"

struct c32h { float x, y; };

__device__ __forceinline__ c32h cmul(c32h a, c32h b) {
  return {a.x * b.x - a.y * b.y, a.x * b.y + a.y * b.x};
}
__device__ __forceinline__ c32h cmulc(c32h a, c32h b) {  // a * conj(b)
  return {a.x * b.x + a.y * b.y, a.y * b.x - a.x * b.y};
}
__device__ __forceinline__ c32h cadd(c32h a, c32h b) {
  return {a.x + b.x, a.y + b.y};
}

#define HB_PER_BLOCK 4   // baselines per 64-thread workgroup

extern "C" __global__ __launch_bounds__(64) void hessianres_kernel(
    const c32h* __restrict__ C,   // (K, T*B, 4) rows [XX,XY,YX,YY]
    const c32h* __restrict__ R,   // (T*B, 2, 2) residual blocks
    const c32h* __restrict__ J,   // (K, N, 2, 2) station Jones
    const int* __restrict__ p_idx,  // (B,)
    const int* __restrict__ q_idx,  // (B,)
    c32h* __restrict__ H,         // (K, 4N, 4N) pre-zeroed
    int K, int N, int B, int T) {
  const int k = blockIdx.y;
  const int bl_raw = blockIdx.x * HB_PER_BLOCK + (threadIdx.x >> 4);
  // no early return: every thread must reach the __syncthreads below
  const bool valid = bl_raw < B;
  const int bl = valid ? bl_raw : 0;
  const int e = threadIdx.x & 15;        // element of the 4x4 block
  const int row = e >> 2, col = e & 3;
  const int i = row >> 1, a = row & 1;   // off rows (i,a)
  const int j = col >> 1, c = col & 1;   // off cols (j,c)
  const int p = p_idx[bl], q = q_idx[bl];
  const float inv = 1.f / ((float)B * (float)T);

  // G_p = Jp^H Jp, G_q = Jq^H Jq (2x2, t-independent)
  c32h Jp[2][2], Jq[2][2];
#pragma unroll
  for (int r = 0; r < 2; ++r)
#pragma unroll
    for (int cc = 0; cc < 2; ++cc) {
      Jp[r][cc] = J[((long)k * N + p) * 4 + r * 2 + cc];
      Jq[r][cc] = J[((long)k * N + q) * 4 + r * 2 + cc];
    }
  c32h Gp[2][2], Gq[2][2];
#pragma unroll
  for (int r = 0; r < 2; ++r)
#pragma unroll
    for (int cc = 0; cc < 2; ++cc) {
      // G[r][cc] = sum_m conj(Jm r) * J[m][cc]
      c32h gp = {0.f, 0.f}, gq = {0.f, 0.f};
#pragma unroll
      for (int m = 0; m < 2; ++m) {
        gp = cadd(gp, cmulc(Jp[m][cc], Jp[m][r]));
        gq = cadd(gq, cmulc(Jq[m][cc], Jq[m][r]));
      }
      Gp[r][cc] = gp;
      Gq[r][cc] = gq;
    }

  c32h off = {0.f, 0.f};
  c32h d1 = {0.f, 0.f};   // (Ci Gq Ci^H)[j][i]
  c32h d2 = {0.f, 0.f};   // (Ci^H Gp Ci)[j][i]
  for (int t = 0; t < T; ++t) {
    const long s = (long)t * B + bl;
    // column-major 2x2: Ci[r][cc] = C4[cc*2 + r]
    c32h Ci[2][2];
    const c32h* C4 = &C[((long)k * T * B + s) * 4];
    Ci[0][0] = C4[0]; Ci[1][0] = C4[1];
    Ci[0][1] = C4[2]; Ci[1][1] = C4[3];
    const c32h* Rs = &R[s * 4];
    // off element: -conj(Ci[i][j]) * Res[a][c]
    {
      c32h cc_ = {Ci[i][j].x, -Ci[i][j].y};
      c32h t1 = cmul(cc_, Rs[a * 2 + c]);
      off.x -= t1.x;
      off.y -= t1.y;
    }
    // d1[j][i] = sum_{m,n} Ci[j][m] Gq[m][n] conj(Ci[i][n])
    // d2[j][i] = sum_{m,n} conj(Ci[m][j]) Gp[m][n] Ci[n][i]
#pragma unroll
    for (int m = 0; m < 2; ++m)
#pragma unroll
      for (int n = 0; n < 2; ++n) {
        d1 = cadd(d1, cmulc(cmul(Ci[j][m], Gq[m][n]), Ci[i][n]));
        c32h cmj = {Ci[m][j].x, -Ci[m][j].y};
        d2 = cadd(d2, cmul(cmul(cmj, Gp[m][n]), Ci[n][i]));
      }
  }

  const long ld = 4L * N;
  c32h* Hk = H + (long)k * ld * ld;
  // off (p,q) and its hermitian mirror (q,p): mirror[(row,col)] =
  // conj(off[(col,row)]) — exchange within the 16-thread group via DPP-
  // free LDS (tiny)
  __shared__ c32h offs[HB_PER_BLOCK][16];
  offs[threadIdx.x >> 4][e] = off;
  __syncthreads();
  const c32h offT = offs[threadIdx.x >> 4][col * 4 + row];
  if (!valid) return;
  Hk[(4L * p + row) * ld + 4L * q + col] = {off.x * inv, off.y * inv};
  Hk[(4L * q + row) * ld + 4L * p + col] = {offT.x * inv, -offT.y * inv};
  // diagonal blocks: nonzero only where the I2 factor hits (a == c)
  if (a == c) {
    float* dst1 = reinterpret_cast<float*>(
        &Hk[(4L * p + row) * ld + 4L * p + col]);
    atomicAdd(dst1 + 0, d1.x * inv);
    atomicAdd(dst1 + 1, d1.y * inv);
    float* dst2 = reinterpret_cast<float*>(
        &Hk[(4L * q + row) * ld + 4L * q + col]);
    atomicAdd(dst2 + 0, d2.x * inv);
    atomicAdd(dst2 + 1, d2.y * inv);
  }
}
