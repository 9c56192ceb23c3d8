// Coherency (model visibility) prediction — N8 of SURVEY.md §2.2.
//
// Reference: `calibration_tools.skytocoherencies[_torch|_uvw]`
// (calibration_tools.py:215-464) — a python loop over sources. The
// torch rewrite (radio/coherency.py) is ~10 kernels per source-chunk
// per cluster; this kernel does the WHOLE prediction in ONE launch:
// grid (T-tiles × clusters), each thread owns one sample t and loops
// the cluster's sources from an LDS-staged table, accumulating the
// complex double sum  Σ_s amp_s(u,v,w) · e^{i φ_s}  in registers.
//
// Per-source table row (11 doubles, precomputed host-side):
//   [l, m, n,  flux,  gflag,  gu1, gv1, gw1,  gu2, gv2, gw2]
// φ = u·l + v·m + w·n (uvw pre-scaled by 2π f / c);
// smearing: amp *= |sinc(φ · fdelta/2 / π)| (fdelta = bw/f, 0 = off);
// Gaussian: amp *= π/2 · exp(−(uut² + vvt²)) with uut = gu1·u+gv1·v+gw1·w,
// vvt = gu2·u+gv2·v+gw2·w (the projected-envelope rotation is linear in
// uvw, so it collapses to 6 coefficients per source).

#include "common.h"

#define COH_CHUNK 128   // sources staged per LDS pass (128*11*8 = 11 KB)

extern "C" __global__ __launch_bounds__(256) void coherency_kernel(
    const double* __restrict__ UVW,  // (T, 3) pre-scaled
    const double* __restrict__ SRC,  // (Stot, 11)
    const int* __restrict__ OFF,     // (K+1) cluster offsets into SRC
    float* __restrict__ C,           // (K, T, 4) interleaved complex64
    double fdelta, int T, int K) {
  __shared__ double tab[COH_CHUNK * 11];
  const int t = blockIdx.x * blockDim.x + threadIdx.x;
  const int k = blockIdx.y;
  const int s_lo = OFF[k], s_hi = OFF[k + 1];

  double u = 0.0, v = 0.0, w = 0.0;
  if (t < T) {
    u = UVW[3 * t + 0];
    v = UVW[3 * t + 1];
    w = UVW[3 * t + 2];
  }
  double acc_re = 0.0, acc_im = 0.0;
  for (int c0 = s_lo; c0 < s_hi; c0 += COH_CHUNK) {
    const int cn = min(COH_CHUNK, s_hi - c0);
    for (int i = threadIdx.x; i < cn * 11; i += blockDim.x)
      tab[i] = SRC[(long)(c0) * 11 + i];
    __syncthreads();
    if (t < T) {
      for (int s = 0; s < cn; ++s) {
        const double* r = &tab[s * 11];
        const double ph = u * r[0] + v * r[1] + w * r[2];
        double amp = r[3];
        if (fdelta > 0.0) {
          const double x = ph * (0.5 * fdelta);
          // |sinc(x/pi)| in the numpy convention = |sin(x)/x|
          amp *= (fabs(x) < 1e-12) ? 1.0 : fabs(sin(x) / x);
        }
        if (r[4] != 0.0) {
          const double uut = r[5] * u + r[6] * v + r[7] * w;
          const double vvt = r[8] * u + r[9] * v + r[10] * w;
          amp *= 0.5 * M_PI * exp(-(uut * uut + vvt * vvt));
        }
        acc_re += amp * cos(ph);
        acc_im += amp * sin(ph);
      }
    }
    __syncthreads();
  }
  if (t < T) {
    const long o = ((long)k * T + t) * 8;   // 4 complex = 8 floats
    C[o + 0] = (float)acc_re;               // XX
    C[o + 1] = (float)acc_im;
    C[o + 2] = 0.f;                         // XY
    C[o + 3] = 0.f;
    C[o + 4] = 0.f;                         // YX
    C[o + 5] = 0.f;
    C[o + 6] = (float)acc_re;               // YY
    C[o + 7] = (float)acc_im;
  }
}
