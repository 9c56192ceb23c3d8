// Fused ALS-sweep contribution kernel for the calibration solver
// (radio/solver.py::_solve_sweeps). One thread per visibility sample
// computes the per-direction Jones products
//     A_p^k = C_k J_q^{kH},   A_q^k = C_k^H J_p^{kH}
// and the per-sample normal-equation contributions
//     rhs_p = V W_p^H, rhs_q = V^H W_q^H, nm_p = W_p W_p^H, nm_q = ...
// (W = the A^k stacked over directions, contraction over the trailing
// dim of size 2), writing them in the concatenated [p-side; q-side]
// layout the gather+sum reduction consumes. This replaces ~100 broadcast
// elementwise kernels per sweep (the pure-torch complex 2x2 path) with
// ONE launch — the radio env step is dispatch-bound.

#include "common.h"

#define KMAX 8

typedef struct {
  float x, y;
} c32;

__device__ __forceinline__ c32 cmul(c32 a, c32 b) {
  return {a.x * b.x - a.y * b.y, a.x * b.y + a.y * b.x};
}
__device__ __forceinline__ c32 cmulj(c32 a, c32 b) {  // a * conj(b)
  return {a.x * b.x + a.y * b.y, a.y * b.x - a.x * b.y};
}
__device__ __forceinline__ c32 cjmul(c32 a, c32 b) {  // conj(a) * b
  return {a.x * b.x + a.y * b.y, a.x * b.y - a.y * b.x};
}
__device__ __forceinline__ c32 cadd(c32 a, c32 b) {
  return {a.x + b.x, a.y + b.y};
}

extern "C" __global__ void als_sweep_kernel(
    const c32* __restrict__ C22,   // (F,K,T,B,2,2)
    const c32* __restrict__ V22,   // (F,T,B,2,2)
    const c32* __restrict__ J,     // (F,Ts,K,N,2,2)
    const int* __restrict__ p_idx, // (B,)
    const int* __restrict__ q_idx, // (B,)
    const int* __restrict__ t_int, // (T,)
    c32* __restrict__ rhs_cat,     // (F, 2*2K, 2*T*B) — entry-major so
    c32* __restrict__ nm_cat,      // (F, 2K*2K, 2*T*B)  writes coalesce
    int F, int K, int T, int B, int N, int Ts) {
  const long s = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long TB = (long)T * B;
  if (s >= (long)F * TB) return;
  const int f = (int)(s / TB);
  const long tb = s % TB;
  const int t = (int)(tb / B);
  const int b = (int)(tb % B);
  const int ti = t_int[t];
  const int pp = p_idx[b];
  const int qq = q_idx[b];
  const int K2 = 2 * K;

  // load V (2x2)
  c32 V[2][2];
  {
    const c32* vp = V22 + (((long)f * T + t) * B + b) * 4;
    V[0][0] = vp[0]; V[0][1] = vp[1]; V[1][0] = vp[2]; V[1][1] = vp[3];
  }

  // per-direction A_p, A_q (2x2 each)
  c32 Ap[KMAX][2][2], Aq[KMAX][2][2];
#pragma unroll
  for (int k = 0; k < KMAX; ++k) {
    if (k >= K) break;
    c32 Cm[2][2], Jq[2][2], Jp[2][2];
    {
      const c32* cp = C22 + ((((long)f * K + k) * T + t) * B + b) * 4;
      Cm[0][0] = cp[0]; Cm[0][1] = cp[1]; Cm[1][0] = cp[2]; Cm[1][1] = cp[3];
      const c32* jb = J + ((((long)f * Ts + ti) * K + k) * N) * 4;
      const c32* jq = jb + (long)qq * 4;
      Jq[0][0] = jq[0]; Jq[0][1] = jq[1]; Jq[1][0] = jq[2]; Jq[1][1] = jq[3];
      const c32* jp = jb + (long)pp * 4;
      Jp[0][0] = jp[0]; Jp[0][1] = jp[1]; Jp[1][0] = jp[2]; Jp[1][1] = jp[3];
    }
    // Ap = Cm @ Jq^H : Ap[i][j] = sum_m Cm[i][m] * conj(Jq[j][m])
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j)
        Ap[k][i][j] = cadd(cmulj(Cm[i][0], Jq[j][0]),
                           cmulj(Cm[i][1], Jq[j][1]));
    // Aq = Cm^H @ Jp^H : Aq[i][j] = sum_m conj(Cm[m][i]) * conj(Jp[j][m])
    //                             = conj( sum_m Cm[m][i] * Jp[j][m] )
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        c32 acc = cadd(cmul(Cm[0][i], Jp[j][0]), cmul(Cm[1][i], Jp[j][1]));
        Aq[k][i][j] = {acc.x, -acc.y};
      }
  }

  // output columns: p side at col tb, q side at col TB + tb; entry e of
  // sample col lives at ((f*X + e) * 2TB + col) so adjacent lanes write
  // adjacent addresses (coalesced)
  const long TB2 = 2 * TB;
  c32* rp = rhs_cat + (long)f * (2 * K2) * TB2 + tb;
  c32* rq = rp + TB;
  // rhs_p[i][c=2k+j] = sum_t V[i][t] conj(W_p[c][t]),  W_p[2k+j][t]=Ap[k][j][t]
#pragma unroll
  for (int k = 0; k < KMAX; ++k) {
    if (k >= K) break;
#pragma unroll
    for (int j = 0; j < 2; ++j) {
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        c32 vp = cadd(cmulj(V[i][0], Ap[k][j][0]),
                      cmulj(V[i][1], Ap[k][j][1]));
        rp[(long)(i * K2 + 2 * k + j) * TB2] = vp;
        // rhs_q uses V^H: (V^H)[i][t] = conj(V[t][i])
        // rhs_q[i][c] = sum_t conj(V[t][i]) * conj(Aq[c][t])
        c32 vq = cadd(cmulj((c32){V[0][i].x, -V[0][i].y}, Aq[k][j][0]),
                      cmulj((c32){V[1][i].x, -V[1][i].y}, Aq[k][j][1]));
        rq[(long)(i * K2 + 2 * k + j) * TB2] = vq;
      }
    }
  }
  // nm[r][c] = sum_t W[r][t] conj(W[c][t]) — Hermitian, compute full
  c32* np_ = nm_cat + (long)f * (K2 * K2) * TB2 + tb;
  c32* nq_ = np_ + TB;
#pragma unroll
  for (int kr = 0; kr < KMAX; ++kr) {
    if (kr >= K) break;
#pragma unroll
    for (int jr = 0; jr < 2; ++jr) {
      const int r = 2 * kr + jr;
#pragma unroll
      for (int kc = 0; kc < KMAX; ++kc) {
        if (kc >= K) break;
#pragma unroll
        for (int jc = 0; jc < 2; ++jc) {
          const int c = 2 * kc + jc;
          np_[(long)(r * K2 + c) * TB2] =
              cadd(cmulj(Ap[kr][jr][0], Ap[kc][jc][0]),
                   cmulj(Ap[kr][jr][1], Ap[kc][jc][1]));
          nq_[(long)(r * K2 + c) * TB2] =
              cadd(cmulj(Aq[kr][jr][0], Aq[kc][jc][0]),
                   cmulj(Aq[kr][jr][1], Aq[kc][jc][1]));
        }
      }
    }
  }
}
