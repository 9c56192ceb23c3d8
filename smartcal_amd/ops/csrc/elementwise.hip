// Fused elementwise kernels: tanh-Gaussian sampling + log-prob (N3) and the
// flat-pool Adam step. One pass each — the reference composes ~10 torch ops
// for the sampler (enet_sac.py:446-466) and torch Adam runs ~6 kernels.

#include "common.h"

#define LOG_SQRT_2PI 0.9189385332046727f
#define REPARAM_NOISE 1e-6f

// action = M*tanh(mu + sigma*eps); per-row logprob summed over action dim.
// One workgroup per batch row (action dims are tiny: 2..128).
extern "C" __global__ void tanh_gauss_fwd_kernel(
    const float* __restrict__ MU, const float* __restrict__ LOGSIG,
    const float* __restrict__ EPS, float* __restrict__ ACT,
    float* __restrict__ LOGP, float* __restrict__ AT, float max_action,
    int B, int A) {
  const int row = blockIdx.x;
  const int lane = threadIdx.x;  // blockDim.x == 64 (one wave)
  float lp = 0.f;
  for (int c = lane; c < A; c += WAVE) {
    const long i = (long)row * A + c;
    const float ls = LOGSIG[i];
    const float sig = expf(ls);
    const float e = EPS[i];
    const float z = MU[i] + sig * e;
    const float at = tanhf(z);
    AT[i] = at;
    ACT[i] = at * max_action;
    lp += -0.5f * e * e - ls - LOG_SQRT_2PI
          - logf(max_action * (1.f - at * at) + REPARAM_NOISE);
  }
  lp = wave_sum(lp);
  if (lane == 0) LOGP[row] = lp;
}

// Backward for the reparameterized path.
// dmu = dact*M*(1-at^2) + dlogp * dlp_dz
// dlogsig = dact*M*(1-at^2)*sig*eps + dlogp * (-1 + dlp_dz*sig*eps)
// dlp_dz = 2*M*at*(1-at^2) / (M*(1-at^2)+1e-6)
extern "C" __global__ void tanh_gauss_bwd_kernel(
    const float* __restrict__ DACT, const float* __restrict__ DLOGP,
    const float* __restrict__ LOGSIG, const float* __restrict__ EPS,
    const float* __restrict__ AT, float* __restrict__ DMU,
    float* __restrict__ DLOGSIG, float max_action, int B, int A) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= (long)B * A) return;
  const int row = i / A;
  const float at = AT[i];
  const float one_m = 1.f - at * at;
  const float sig = expf(LOGSIG[i]);
  const float se = sig * EPS[i];
  const float dl = DLOGP[row];
  const float da = DACT[i];
  const float dlp_dz = 2.f * max_action * at * one_m
                       / (max_action * one_m + REPARAM_NOISE);
  const float dz = da * max_action * one_m + dl * dlp_dz;
  DMU[i] = dz;
  DLOGSIG[i] = dz * se - dl;
}

// Fused Adam over a flat parameter pool (bias-corrected, torch defaults).
// The step counter lives on the DEVICE (tptr) so the whole optimizer step
// is hipGraph-capturable: block 0 thread 0 bumps it, every block computes
// the bias corrections from the pre-bump value.
extern "C" __global__ void fused_adam_kernel(
    float* __restrict__ P, const float* __restrict__ G,
    float* __restrict__ M, float* __restrict__ V, float* __restrict__ tptr,
    float lr, float b1, float b2, float eps, long n) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const float t = tptr[0] + 1.f;
  if (i >= n) return;
  const float bc1 = 1.f - powf(b1, t);
  const float bc2 = 1.f - powf(b2, t);
  const float g = G[i];
  const float m = b1 * M[i] + (1.f - b1) * g;
  const float v = b2 * V[i] + (1.f - b2) * g * g;
  M[i] = m;
  V[i] = v;
  const float mhat = m / bc1;
  const float vhat = v / bc2;
  P[i] -= lr * mhat / (sqrtf(vhat) + eps);
}

// separate tiny bump kernel so the ordering vs fused_adam_kernel is safe
extern "C" __global__ void adam_bump_kernel(float* __restrict__ tptr) {
  if (threadIdx.x == 0 && blockIdx.x == 0) tptr[0] += 1.f;
}
