// Fused gather + segment-sum for the ALS solver's per-(interval,
// station) reduction (radio/solver.py:_solve_sweeps). The torch
// composition materializes the full gathered copy
// (in[:, :, gidx] — F·X·2TB complex, ~260 MB per sweep at LOFAR scale)
// then reduces; this kernel reads the scattered columns once and sums
// in registers: out[f, x, g] = Σ_c in[f, x, gidx[g, c]].
//
// Layout: in (F, X, Scols) complex64, gidx (G, Cnt) int64 (torch
// argsort output), out (F, X, G). One wave per (f, x, g-tile): lanes
// split the Cnt reduction, DPP wave-sum combines.

#include "common.h"

struct c32g { float x, y; };

extern "C" __global__ __launch_bounds__(256) void gather_sum_kernel(
    const c32g* __restrict__ IN, const long* __restrict__ GIDX,
    c32g* __restrict__ OUT, int F, int X, long Scols, int G, int Cnt) {
  // block = 256 threads = 4 waves; each wave owns one output element
  const long flat = (long)blockIdx.x * 4 + (threadIdx.x >> 6);
  const int lane = threadIdx.x & 63;
  const long total = (long)F * X * G;
  if (flat >= total) return;
  const int g = flat % G;
  const long fx = flat / G;          // f * X + x
  const c32g* row = IN + fx * Scols;
  const long* gi = GIDX + (long)g * Cnt;
  float sre = 0.f, sim = 0.f;
  for (int c = lane; c < Cnt; c += WAVE) {
    const c32g v = row[gi[c]];
    sre += v.x;
    sim += v.y;
  }
  sre = wave_sum(sre);
  sim = wave_sum(sim);
  if (lane == 0) OUT[fx * G + g] = {sre, sim};
}
