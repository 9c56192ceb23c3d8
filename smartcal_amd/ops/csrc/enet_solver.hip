// Batched elastic-net L-BFGS solver + influence/eigen/reward kernels (the
// ENetEnv.step() hot path).
//
// The reference runs the env's inner optimization as ~600 closure
// evaluations of a python L-BFGS (reference elasticnet/enetenv.py:94-114 +
// lbfgsnew.py), then N inverse-Hessian-vector products and an
// eigendecomposition — thousands of microscopic kernel launches per env
// step on a GPU. MI355X-native layout: ONE workgroup (one 64-lane wave) per
// environment runs the whole 20-epoch strong-Wolfe L-BFGS in-kernel with
// the problem (A, y, x, gradient, curvature history) resident in LDS and
// every dot product a wave shfl reduction; a second kernel applies the
// two-loop recursion to the matrix RHS (-2 A^T), forms the influence matrix
// B = A H^{-1} (-2 A^T), runs a cyclic-Jacobi eigensolver on it and emits
// the observation eigenvalues + reward. Batched over environments via
// blockIdx.x, so vectorized-env rollouts run hundreds of solves in one
// launch.
//
// Size limits: N, M <= 32 (bench config N=M=20), history <= 7.
// Numerics match smartcal_amd.ops.enet.lbfgs_solve_reference /
// influence_eigs_reference (the CPU oracle) to fp32 tolerance.

#include "common.h"

#define HMAX 7
#define TOL_GRAD 1e-7f
#define TOL_CHANGE 1e-9f
#define NMAX 32

struct LdsLayout {
  float* __restrict__ A;      // N*M
  float* __restrict__ y;      // N
  float* __restrict__ x;      // M
  float* __restrict__ g;      // M
  float* __restrict__ gprev;  // M
  float* __restrict__ d;      // M
  float* __restrict__ x0;     // M
  float* __restrict__ r;      // N
  float* __restrict__ S;      // HMAX*M
  float* __restrict__ Yh;     // HMAX*M
  float* __restrict__ bg0;    // M (bracket grad low)
  float* __restrict__ bg1;    // M (bracket grad high)
  float* __restrict__ ro;     // HMAX (uniform scalars; LDS to avoid scratch spills)
  float* __restrict__ al;     // HMAX
  float* __restrict__ xprev;  // M (previous-epoch x snapshot)
};

// Per-lane register cache of A: row[j] = A[lane][j] (for the forward
// matvec, lanes < N) and col[i] = A[i][lane] (for the A^T r gradient,
// lanes < M). A is constant for the whole solve; keeping it in VGPRs
// and front-loading x/r turns each closure evaluation into two LDS
// waits + pure VALU FMA chains (the interleaved LDS-read form stalled
// on an lgkmcnt wait per FMA).
struct RegA {
  float row[32];
  float col[32];
};

// residual + loss + gradient of ||y-Ax||^2 + rho1||x||^2 + rho2||x||_1
__device__ __forceinline__ static float eval_loss_grad(
    const LdsLayout& L, const RegA& Ar, int N, int M, float rho1,
    float rho2) {
  const int lane = threadIdx.x;
  float xr[32];
#pragma unroll
  for (int j = 0; j < 32; ++j) xr[j] = (j < M) ? L.x[j] : 0.f;
  float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
#pragma unroll
  for (int j = 0; j < 32; j += 4) {
    a0 += Ar.row[j + 0] * xr[j + 0];
    a1 += Ar.row[j + 1] * xr[j + 1];
    a2 += Ar.row[j + 2] * xr[j + 2];
    a3 += Ar.row[j + 3] * xr[j + 3];
  }
  const float ri = (lane < N) ? L.y[lane] - ((a0 + a1) + (a2 + a3)) : 0.f;
  if (lane < N) L.r[lane] = ri;
  const float xl = (lane < M) ? L.x[lane] : 0.f;
  float loss = wave_sum(ri * ri + rho1 * xl * xl + rho2 * fabsf(xl));
  __builtin_amdgcn_s_barrier();  // r[] visible (single wave: lockstep, but
                                 // keep an explicit ordering point)
  float rr[32];
#pragma unroll
  for (int i = 0; i < 32; ++i) rr[i] = (i < N) ? L.r[i] : 0.f;
  float b0 = 0.f, b1 = 0.f, b2 = 0.f, b3 = 0.f;
#pragma unroll
  for (int i = 0; i < 32; i += 4) {
    b0 += Ar.col[i + 0] * rr[i + 0];
    b1 += Ar.col[i + 1] * rr[i + 1];
    b2 += Ar.col[i + 2] * rr[i + 2];
    b3 += Ar.col[i + 3] * rr[i + 3];
  }
  if (lane < M) {
    const float atr = (b0 + b1) + (b2 + b3);
    const float sgn = (xl > 0.f) ? 1.f : (xl < 0.f ? -1.f : 0.f);
    L.g[lane] = -2.f * atr + 2.f * rho1 * xl + rho2 * sgn;
  }
  return loss;
}

__device__ static float lds_dot(const float* a, const float* b, int n) {
  const int lane = threadIdx.x;
  float v = (lane < n) ? a[lane] * b[lane] : 0.f;
  return wave_sum(v);
}

__device__ static float lds_absmax(const float* a, int n) {
  const int lane = threadIdx.x;
  float v = (lane < n) ? fabsf(a[lane]) : 0.f;
  return wave_max(v);
}

// phi(t) = f(x0 + t d); leaves x at the evaluated point, g = grad there.
__device__ static float ls_eval(const LdsLayout& L, const RegA& Ar,
                                int N, int M,
                                float rho1, float rho2, float t,
                                float* gtd_out) {
  const int lane = threadIdx.x;
  if (lane < M) {
    float xd = t * L.d[lane];
    L.x[lane] = L.x0[lane] + (isfinite(xd) ? xd : 0.f);
  }
  float f = eval_loss_grad(L, Ar, N, M, rho1, rho2);
  *gtd_out = lds_dot(L.g, L.d, M);
  return f;
}

__device__ static float cubic_interp(float x1, float f1, float g1, float x2,
                                     float f2, float g2, float lo, float hi) {
  // NaN/degenerate guards mirroring the reference line search's checks
  // (`lbfgsnew.py:556,624,673`): fall back to the bisection midpoint
  if (fabsf(x1 - x2) < 1e-20f || !isfinite(f1) || !isfinite(f2) ||
      !isfinite(g1) || !isfinite(g2))
    return 0.5f * (lo + hi);
  float d1 = g1 + g2 - 3.f * (f1 - f2) / (x1 - x2);
  float d2sq = d1 * d1 - g1 * g2;
  if (isfinite(d2sq) && d2sq >= 0.f) {
    float d2 = sqrtf(d2sq);
    float mp;
    if (x1 <= x2)
      mp = x2 - (x2 - x1) * ((g2 + d2 - d1) / (g2 - g1 + 2.f * d2));
    else
      mp = x1 - (x1 - x2) * ((g1 + d2 - d1) / (g1 - g2 + 2.f * d2));
    if (!isfinite(mp)) return 0.5f * (lo + hi);
    return fminf(fmaxf(mp, lo), hi);
  }
  return 0.5f * (lo + hi);
}

// Strong-Wolfe line search (same algorithm as optim/lbfgs.py::_strong_wolfe).
// Returns f at the accepted point; x and g hold that point on exit; *t_io
// the accepted step.
__device__ static float strong_wolfe(const LdsLayout& L, const RegA& Ar,
                                     int N, int M,
                                     float rho1, float rho2, float f0,
                                     float gtd0, float* t_io) {
  const int lane = threadIdx.x;
  const float c1 = 1e-4f, c2 = 0.9f;
  const int max_ls = 25;
  float t = *t_io;
  float d_norm = lds_absmax(L.d, M);

  // bracket state: index 0/1
  float br_t[2], br_f[2], br_gtd[2];
  float* br_g[2] = {L.bg0, L.bg1};
  bool done = false;
  int ls_iter = 0;

  float t_prev = 0.f, f_prev = f0, gtd_prev = gtd0;
  // bg0 starts as g(0)
  if (lane < M) L.bg0[lane] = L.g[lane];

  float f_new, gtd_new;
  bool bracketed = false;
  while (ls_iter < max_ls) {
    f_new = ls_eval(L, Ar, N, M, rho1, rho2, t, &gtd_new);
    if (!isfinite(f_new) || !isfinite(gtd_new) || !isfinite(t)) {
      // blown-up step: restore the start point (reference NaN guard)
      t = 0.f;
      f_new = ls_eval(L, Ar, N, M, rho1, rho2, 0.f, &gtd_new);
      *t_io = 0.f;
      return f_new;
    }
    if (f_new > (f0 + c1 * t * gtd0) || (ls_iter > 0 && f_new >= f_prev)) {
      br_t[0] = t_prev; br_f[0] = f_prev; br_gtd[0] = gtd_prev;
      // bg0 already holds g(t_prev)
      br_t[1] = t; br_f[1] = f_new; br_gtd[1] = gtd_new;
      if (lane < M) L.bg1[lane] = L.g[lane];
      bracketed = true;
      break;
    }
    if (fabsf(gtd_new) <= -c2 * gtd0) {
      br_t[0] = t; br_f[0] = f_new;
      if (lane < M) L.bg0[lane] = L.g[lane];
      done = true; bracketed = true;
      break;
    }
    if (gtd_new >= 0.f) {
      br_t[0] = t_prev; br_f[0] = f_prev; br_gtd[0] = gtd_prev;
      br_t[1] = t; br_f[1] = f_new; br_gtd[1] = gtd_new;
      if (lane < M) L.bg1[lane] = L.g[lane];
      bracketed = true;
      break;
    }
    float min_step = t + 0.01f * (t - t_prev);
    float max_step = t * 10.f;
    float t_next = cubic_interp(t_prev, f_prev, gtd_prev, t, f_new, gtd_new,
                                min_step, max_step);
    t_prev = t; f_prev = f_new; gtd_prev = gtd_new;
    if (lane < M) L.bg0[lane] = L.g[lane];
    t = t_next;
    ++ls_iter;
  }
  if (!bracketed) {
    br_t[0] = 0.f; br_f[0] = f0; br_gtd[0] = gtd0;
    // bg0 holds g at last eval — for the unbracketed fallback use t side
    br_t[1] = t; br_f[1] = f_new; br_gtd[1] = gtd_new;
    if (lane < M) L.bg1[lane] = L.g[lane];
  }

  int low = (br_f[0] <= br_f[1]) ? 0 : 1;
  int high = 1 - low;
  bool insuf = false;
  while (!done && ls_iter < max_ls) {
    if (fabsf(br_t[1] - br_t[0]) * d_norm < 1e-10f) break;
    t = cubic_interp(br_t[0], br_f[0], br_gtd[0], br_t[1], br_f[1],
                     br_gtd[1], fminf(br_t[0], br_t[1]),
                     fmaxf(br_t[0], br_t[1]));
    float bmax = fmaxf(br_t[0], br_t[1]);
    float bmin = fminf(br_t[0], br_t[1]);
    float eps_b = 0.1f * (bmax - bmin);
    if (fminf(bmax - t, t - bmin) < eps_b) {
      if (insuf || t >= bmax || t <= bmin) {
        t = (fabsf(t - bmax) < fabsf(t - bmin)) ? bmax - eps_b
                                                : bmin + eps_b;
        insuf = false;
      } else {
        insuf = true;
      }
    } else {
      insuf = false;
    }
    f_new = ls_eval(L, Ar, N, M, rho1, rho2, t, &gtd_new);
    if (!isfinite(f_new) || !isfinite(gtd_new)) {
      t = 0.f;
      f_new = ls_eval(L, Ar, N, M, rho1, rho2, 0.f, &gtd_new);
      *t_io = 0.f;
      return f_new;
    }
    if (f_new > (f0 + c1 * t * gtd0) || f_new >= br_f[low]) {
      br_t[high] = t; br_f[high] = f_new; br_gtd[high] = gtd_new;
      if (lane < M) br_g[high][lane] = L.g[lane];
      low = (br_f[0] <= br_f[1]) ? 0 : 1;
      high = 1 - low;
    } else {
      if (fabsf(gtd_new) <= -c2 * gtd0) {
        done = true;
      } else if (gtd_new * (br_t[high] - br_t[low]) >= 0.f) {
        br_t[high] = br_t[low]; br_f[high] = br_f[low];
        br_gtd[high] = br_gtd[low];
        if (lane < M) br_g[high][lane] = br_g[low][lane];
      }
      br_t[low] = t; br_f[low] = f_new; br_gtd[low] = gtd_new;
      if (lane < M) br_g[low][lane] = L.g[lane];
    }
    ++ls_iter;
  }

  if (!done) {
    t = br_t[low];
    f_new = br_f[low];
    if (lane < M) {
      L.g[lane] = br_g[low][lane];
      float xd = t * L.d[lane];
      L.x[lane] = L.x0[lane] + (isfinite(xd) ? xd : 0.f);
    }
  }
  // done==true: x and g already hold the accepted point
  *t_io = t;
  return f_new;
}

extern "C" __global__ __launch_bounds__(64) void enet_lbfgs_solve_kernel(
    const float* __restrict__ Ag,   // (E, N, M)
    const float* __restrict__ yg,   // (E, N)
    const float* __restrict__ rhog, // (E, 2)
    float* __restrict__ xg,         // (E, M)
    float* __restrict__ Yg,         // (E, HMAX, M) oldest first
    float* __restrict__ Sg,         // (E, HMAX, M)
    int* __restrict__ nhistg,       // (E,)
    int E, int N, int M, int epochs, int max_iter, int history) {
  extern __shared__ float smem[];
  const int env = blockIdx.x;
  const int lane = threadIdx.x;
  if (history > HMAX) history = HMAX;

  LdsLayout L;
  float* p = smem;
  L.A = p; p += N * M;
  L.y = p; p += N;
  L.x = p; p += M;
  L.g = p; p += M;
  L.gprev = p; p += M;
  L.d = p; p += M;
  L.x0 = p; p += M;
  L.r = p; p += N;
  L.S = p; p += HMAX * M;
  L.Yh = p; p += HMAX * M;
  L.bg0 = p; p += M;
  L.bg1 = p; p += M;
  L.ro = p; p += HMAX;
  L.al = p; p += HMAX;
  L.xprev = p; p += M;

  for (int i = lane; i < N * M; i += WAVE) L.A[i] = Ag[(long)env * N * M + i];
  __builtin_amdgcn_s_barrier();
  // per-lane register cache of A (row for the matvec, column for A^T r)
  RegA Ar;
#pragma unroll
  for (int j = 0; j < 32; ++j)
    Ar.row[j] = (lane < N && j < M) ? L.A[lane * M + j] : 0.f;
#pragma unroll
  for (int i = 0; i < 32; ++i)
    Ar.col[i] = (lane < M && i < N) ? L.A[i * M + lane] : 0.f;
  if (lane < N) L.y[lane] = yg[(long)env * N + lane];
  if (lane < M) {
    L.x[lane] = 0.f;
    L.d[lane] = 0.f;
    L.gprev[lane] = 0.f;
  }
  const float rho1 = rhog[env * 2 + 0];
  const float rho2 = rhog[env * 2 + 1];

  int nhist = 0;
  int n_iter = 0;
  float t = 1.f, H_diag = 1.f, prev_loss = 0.f;
  float prev_epoch_loss = 1e30f;
  if (lane < M) L.xprev[lane] = 1e30f;

  for (int epoch = 0; epoch < epochs; ++epoch) {
    float loss = eval_loss_grad(L, Ar, N, M, rho1, rho2);
    // epoch-level early stop: once the whole optimizer has converged the
    // reference just keeps burning closure evaluations; identical x at
    // fp32, ~3-5x fewer evaluations.
    if (fabsf(loss - prev_epoch_loss) <= 3e-7f * fmaxf(1.f, fabsf(loss)))
      break;
    prev_epoch_loss = loss;
    // second stop: parameters no longer moving between epochs
    {
      float dx = (lane < M) ? fabsf(L.x[lane] - L.xprev[lane]) : 0.f;
      float dmax = wave_max(dx);
      if (lane < M) L.xprev[lane] = L.x[lane];
      if (dmax <= 1e-7f) break;
    }
    if (lds_absmax(L.g, M) <= TOL_GRAD) break;
    bool outer_done = false;
    for (int it = 0; it < max_iter; ++it) {
      ++n_iter;
      // curvature update
      if (n_iter > 1) {
        float yl = 0.f, sl = 0.f;
        if (lane < M) {
          yl = L.g[lane] - L.gprev[lane];
          sl = L.d[lane] * t;
        }
        float ys = wave_sum(yl * sl);
        float ss = wave_sum(sl * sl);
        if (ys > 1e-10f * ss) {
          if (nhist == history) {
            // drop oldest: shift
            for (int h = 1; h < history; ++h) {
              if (lane < M) {
                L.S[(h - 1) * M + lane] = L.S[h * M + lane];
                L.Yh[(h - 1) * M + lane] = L.Yh[h * M + lane];
              }
              if (lane == 0) L.ro[h - 1] = L.ro[h];
            }
            --nhist;
          }
          if (lane < M) {
            L.S[nhist * M + lane] = sl;
            L.Yh[nhist * M + lane] = yl;
          }
          if (lane == 0) L.ro[nhist] = 1.f / ys;
          ++nhist;
          H_diag = ys / wave_sum(yl * yl);
        }
      }
      // direction
      if (nhist == 0) {
        if (lane < M) L.d[lane] = -L.g[lane];
        H_diag = 1.f;
      } else {
        if (lane < M) L.d[lane] = -L.g[lane];
        for (int i = nhist - 1; i >= 0; --i) {
          float ali = L.ro[i] * lds_dot(&L.S[i * M], L.d, M);
          if (lane == 0) L.al[i] = ali;
          if (lane < M) L.d[lane] -= ali * L.Yh[i * M + lane];
        }
        if (lane < M) L.d[lane] *= H_diag;
        for (int i = 0; i < nhist; ++i) {
          float be = L.ro[i] * lds_dot(&L.Yh[i * M], L.d, M);
          if (lane < M) L.d[lane] += (L.al[i] - be) * L.S[i * M + lane];
        }
      }
      {
        float bad = wave_sum((lane < M && !isfinite(L.d[lane])) ? 1.f : 0.f);
        if (bad > 0.f) {
          if (lane < M) L.d[lane] = -L.g[lane];
          nhist = 0;
          H_diag = 1.f;
        }
      }
      if (lane < M) L.gprev[lane] = L.g[lane];
      prev_loss = loss;

      float gtd = lds_dot(L.g, L.d, M);
      // negated form so a NaN gtd also exits instead of entering the search
      if (!(gtd <= -TOL_CHANGE)) { outer_done = true; break; }

      if (n_iter == 1) {
        float gsum = wave_sum((lane < M) ? fabsf(L.g[lane]) : 0.f);
        t = fminf(1.f, 1.f / gsum);
      } else {
        t = 1.f;
      }

      if (lane < M) L.x0[lane] = L.x[lane];
      loss = strong_wolfe(L, Ar, N, M, rho1, rho2, loss, gtd, &t);
      if (!isfinite(loss)) {
        // restore the pre-line-search point and stop (reference
        // behaviour: NaN warning + early exit, lbfgsnew.py:695-714)
        if (lane < M) L.x[lane] = L.x0[lane];
        loss = eval_loss_grad(L, Ar, N, M, rho1, rho2);
        outer_done = true;
        break;
      }

      if (lds_absmax(L.g, M) <= TOL_GRAD) { outer_done = true; break; }
      float step_max = lds_absmax(L.d, M) * fabsf(t);
      if (step_max <= TOL_CHANGE) break;
      if (fabsf(loss - prev_loss) < TOL_CHANGE) break;
    }
    if (outer_done) break;
  }

  // outputs
  if (lane < M) xg[(long)env * M + lane] = L.x[lane];
  for (int h = 0; h < HMAX; ++h) {
    if (lane < M) {
      float yv = (h < nhist) ? L.Yh[h * M + lane] : 0.f;
      float sv = (h < nhist) ? L.S[h * M + lane] : 0.f;
      Yg[((long)env * HMAX + h) * M + lane] = yv;
      Sg[((long)env * HMAX + h) * M + lane] = sv;
    }
  }
  if (lane == 0) nhistg[env] = nhist;
}

// ---------------------------------------------------------------------------
// Influence + eigenvalues + reward.
// mm = H^{-1} (-2 A^T) via two-loop on the matrix; B = A mm (symmetric);
// cyclic Jacobi for eigenvalues; EE = 1 + ev sorted ascending;
// reward = ||y||/||Ax-y|| + min(EE)/max(EE) + penalty.
extern "C" __global__ __launch_bounds__(64) void enet_influence_kernel(
    const float* __restrict__ Ag,   // (E, N, M)
    const float* __restrict__ yg,   // (E, N)
    const float* __restrict__ xg,   // (E, M)
    const float* __restrict__ Yg,   // (E, HMAX, M)
    const float* __restrict__ Sg,   // (E, HMAX, M)
    const int* __restrict__ nhistg, // (E,)
    const float* __restrict__ peng, // (E,)
    const float* __restrict__ rhog, // (E, 2) — rho1 drives the filter
    float* __restrict__ EEg,        // (E, N)
    float* __restrict__ rewardg,    // (E,)
    int E, int N, int M) {
  extern __shared__ float smem[];
  const int env = blockIdx.x;
  const int lane = threadIdx.x;
  const int nh = nhistg[env];

  float* p = smem;
  float* A = p; p += N * M;
  float* Q = p; p += M * N;     // then reused as R
  float* B = p; p += N * N;
  float* Sv = p; p += HMAX * M;
  float* Yv = p; p += HMAX * M;
  float* alv = p; p += HMAX * N;
  float* ev = p; p += N;
  float* xs = p; p += M;
  float* ys_ = p; p += N;
  float* ro = p; p += HMAX;

  for (int i = lane; i < N * M; i += WAVE) A[i] = Ag[(long)env * N * M + i];
  for (int i = lane; i < HMAX * M; i += WAVE) {
    Sv[i] = Sg[(long)env * HMAX * M + i];
    Yv[i] = Yg[(long)env * HMAX * M + i];
  }
  if (lane < M) xs[lane] = xg[(long)env * M + lane];
  if (lane < N) ys_[lane] = yg[(long)env * N + lane];

  // Q column cached per lane in REGISTERS: qc[m] = Q[m][lane] with
  // Q = -2 A^T, i.e. qc[m] = -2 A[lane][m] (this lane's row of A).
  // The two-loop recursion then runs on registers with one front-loaded
  // LDS broadcast per history entry instead of an lgkmcnt wait per FMA.
  __builtin_amdgcn_s_barrier();
  float qc[32];
#pragma unroll
  for (int m = 0; m < 32; ++m)
    qc[m] = (lane < N && m < M) ? -2.f * A[lane * M + m] : 0.f;

  // Degenerate-pair filter AT THE ANALYTIC BAND. The smooth Hessian is
  // 2(A^T A + rho1 I) with ||A||_F = 1, so every genuine pair y = H s
  // satisfies BOTH  ys >= 2 rho1 ss  (monotone L1 subgradient included)
  // AND  yy <= 4 (1+rho1)^2 ss.  Line-search noise violates one side or
  // the other: tiny-ys pairs inflate the two-loop via 1/ys, and pairs
  // whose y is dominated by L1 sign jumps over a minuscule step have
  // ys/ss up to ~3e5 (measured, gpurun_out/explosion_debug.json) —
  // their (I - rho s y^T) factors carry norm 1/cos(y,s) and blow
  // H^{-1}'s spectrum up, producing the huge negative min(EE)/max(EE)
  // rewards the round-2 curve audit caught. Filter: rho1 ss < ys AND
  // yy < 8 (1+rho1)^2 ss (2x margin each side); ro = 0 no-ops a pair in
  // both loops. With every pair rejected H^{-1} = I, so EE stays in
  // [1 - 2 sigma_max^2, 1] — bounded. The CPU oracle
  // (`ops/enet.py::influence_eigs_reference`) applies the same band.
  const float rho1f = rhog[env * 2];
  const float lo = fmaxf(1e-6f, rho1f);
  const float hi = 8.f * (1.f + rho1f) * (1.f + rho1f);
  float ys = 1.f, yy = 1.f;
  for (int i = 0; i < nh; ++i) {
    float ysi = lds_dot(&Yv[i * M], &Sv[i * M], M);
    float ssi = lds_dot(&Sv[i * M], &Sv[i * M], M);
    float yyi = lds_dot(&Yv[i * M], &Yv[i * M], M);
    float r = (ysi > lo * ssi && yyi < hi * ssi) ? 1.f / ysi : 0.f;
    if (lane == 0) ro[i] = r;
    if (r != 0.f) {  // H_diag scale from the newest GOOD pair
      ys = ysi;
      yy = yyi;
    }
  }
  __builtin_amdgcn_s_barrier();

  // two-loop on the register-resident columns; al_r in registers
  float al_r[HMAX];
#pragma unroll
  for (int i = HMAX - 1; i >= 0; --i) {
    if (i >= nh) continue;
    float sv[32], yv[32];
#pragma unroll
    for (int m = 0; m < 32; ++m) {
      sv[m] = (m < M) ? Sv[i * M + m] : 0.f;
      yv[m] = (m < M) ? Yv[i * M + m] : 0.f;
    }
    float sacc = 0.f;
#pragma unroll
    for (int m = 0; m < 32; ++m) sacc += sv[m] * qc[m];
    const float ai = ro[i] * sacc;
    al_r[i] = ai;
#pragma unroll
    for (int m = 0; m < 32; ++m) qc[m] -= yv[m] * ai;
  }
  const float scale = ys / yy;  // 1 when no good pair (init values)
#pragma unroll
  for (int m = 0; m < 32; ++m) qc[m] *= scale;
#pragma unroll
  for (int i = 0; i < HMAX; ++i) {
    if (i >= nh) continue;
    float sv[32], yv[32];
#pragma unroll
    for (int m = 0; m < 32; ++m) {
      sv[m] = (m < M) ? Sv[i * M + m] : 0.f;
      yv[m] = (m < M) ? Yv[i * M + m] : 0.f;
    }
    float yacc = 0.f;
#pragma unroll
    for (int m = 0; m < 32; ++m) yacc += yv[m] * qc[m];
    const float be = ro[i] * yacc;
#pragma unroll
    for (int m = 0; m < 32; ++m) qc[m] += sv[m] * (al_r[i] - be);
  }

  // B = A @ R, symmetrized (R columns live in qc registers)
  for (int r = 0; r < N; ++r) {
    float ar[32];
#pragma unroll
    for (int m = 0; m < 32; ++m) ar[m] = (m < M) ? A[r * M + m] : 0.f;
    float sacc = 0.f;
#pragma unroll
    for (int m = 0; m < 32; ++m) sacc += ar[m] * qc[m];
    if (lane < N) B[r * N + lane] = sacc;
  }
  __builtin_amdgcn_s_barrier();
  for (int r = 0; r < N; ++r)
    if (lane < N && lane > r) {
      float v = 0.5f * (B[r * N + lane] + B[lane * N + r]);
      B[r * N + lane] = v;
      B[lane * N + r] = v;
    }

  // cyclic Jacobi (values only); break on RELATIVE off-diagonal norm
  float dscale = 0.f;
  {
    float dv = (lane < N) ? B[lane * N + lane] : 0.f;
    dscale = wave_sum(dv * dv) + 1e-30f;
  }
  // 8 sweeps of cyclic Jacobi reach ~1e-6 relative eigenvalue
  // accuracy on these 20x20 symmetric B (fp32 floor); the relative
  // off-diagonal break rarely fires before the cap
  for (int sweep = 0; sweep < 8; ++sweep) {
    float off = 0.f;
    for (int pi = 0; pi < N - 1; ++pi)
      if (lane < N && lane > pi) off += B[pi * N + lane] * B[pi * N + lane];
    off = wave_sum(off);
    if (off < 1e-11f * dscale) break;
    for (int pi = 0; pi < N - 1; ++pi) {
      for (int q = pi + 1; q < N; ++q) {
        float apq = B[pi * N + q];
        if (fabsf(apq) < 1e-12f) continue;
        float app = B[pi * N + pi], aqq = B[q * N + q];
        float theta = 0.5f * (aqq - app) / apq;
        float tt = (theta >= 0.f ? 1.f : -1.f)
                   / (fabsf(theta) + sqrtf(theta * theta + 1.f));
        float c = rsqrtf(tt * tt + 1.f);
        float s = tt * c;
        // rows pi, q
        if (lane < N) {
          float bp = B[pi * N + lane], bq = B[q * N + lane];
          B[pi * N + lane] = c * bp - s * bq;
          B[q * N + lane] = s * bp + c * bq;
        }
        // cols pi, q
        if (lane < N) {
          float bp = B[lane * N + pi], bq = B[lane * N + q];
          B[lane * N + pi] = c * bp - s * bq;
          B[lane * N + q] = s * bp + c * bq;
        }
      }
    }
  }
  // Projection onto the analytically feasible interval: the TRUE
  // H^{-1} <= I/(2 rho1) and sigma_max(A) <= ||A||_F = 1 bound
  // eig(B) >= -1/rho1, i.e. EE >= 1 - 1/rho1; eigenvalues below are
  // quasi-Newton approximation artifacts (the rare hint-arm tail).
  const float ee_floor = 1.f - 1.f / fmaxf(rho1f, 1e-6f);
  if (lane < N) ev[lane] = fmaxf(B[lane * N + lane] + 1.f, ee_floor);

  // odd-even transposition sort ascending (N <= 32)
  for (int phase = 0; phase < N; ++phase) {
    int start = phase & 1;
    int i0 = start + 2 * lane;
    float a0 = 0.f, a1 = 0.f;
    bool act = (i0 + 1 < N);
    if (act) { a0 = ev[i0]; a1 = ev[i0 + 1]; }
    if (act && a0 > a1) {
      ev[i0] = a1;
      ev[i0 + 1] = a0;
    }
  }

  if (lane < N) EEg[(long)env * N + lane] = ev[lane];

  // reward
  float ri = 0.f, yl = 0.f;
  if (lane < N) {
    float ax = 0.f;
    for (int m = 0; m < M; ++m) ax += A[lane * M + m] * xs[m];
    ri = ax - ys_[lane];
    yl = ys_[lane];
  }
  float err = sqrtf(wave_sum(ri * ri));
  float ynorm = sqrtf(wave_sum(yl * yl));
  float emin = ev[0], emax = ev[N - 1];
  if (lane == 0)
    rewardg[env] = ynorm / err + emin / emax + peng[env];
}
