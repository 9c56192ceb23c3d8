// Common device helpers for smartcal_amd CDNA4 (gfx950) kernels.
// Wavefront size on CDNA4 is 64 (not 32) — all cross-lane idioms below are
// 64-wide. fp32 MFMA (v_mfma_f32_16x16x4_f32) is exact f32 at the f32 vector
// rate; used for all GEMM-shaped fp32 work (no TF32 on gfx950).
#pragma once

#include <hip/hip_runtime.h>

#define WAVE 64

// DPP lane permute within a 16-lane row: quad_perm XOR-1 / XOR-2, then
// row-rotate by 4 and 8. DPP runs in the VALU pipe (a few cycles) where
// __shfl_xor lowers to ds_bpermute through the LDS pipe (~10x the
// latency, plus an lgkmcnt wait per step).
template <int CTRL>
__device__ __forceinline__ float _dpp_f(float v) {
  union {
    float f;
    int i;
  } x;
  x.f = v;
  x.i = __builtin_amdgcn_update_dpp(0, x.i, CTRL, 0xF, 0xF, true);
  return x.f;
}

// Sum across a 16-lane group (lanes sharing lane>>4); all 16 get the
// total. 4 DPP ops, no LDS traffic.
__device__ __forceinline__ float group16_sum(float v) {
  v += _dpp_f<0xB1>(v);     // quad_perm [1,0,3,2]  (xor 1)
  v += _dpp_f<0x4E>(v);     // quad_perm [2,3,0,1]  (xor 2)
  v += _dpp_f<0x124>(v);    // row_ror:4
  v += _dpp_f<0x128>(v);    // row_ror:8
  return v;
}

// Sum across all 64 lanes of a wave; every lane gets the total.
// 4 DPP ops for the in-row sum + 2 cross-row butterflies.
__device__ __forceinline__ float wave_sum(float v) {
  v = group16_sum(v);
  v += __shfl_xor(v, 16, WAVE);
  v += __shfl_xor(v, 32, WAVE);
  return v;
}

__device__ __forceinline__ float wave_max(float v) {
  v = fmaxf(v, _dpp_f<0xB1>(v));
  v = fmaxf(v, _dpp_f<0x4E>(v));
  v = fmaxf(v, _dpp_f<0x124>(v));
  v = fmaxf(v, _dpp_f<0x128>(v));
  v = fmaxf(v, __shfl_xor(v, 16, WAVE));
  v = fmaxf(v, __shfl_xor(v, 32, WAVE));
  return v;
}

// fp32 MFMA accumulator fragment: v_mfma_f32_16x16x4_f32.
// Lane l supplies A[i = l&15][k = l>>4] and B[k = l>>4][j = l&15];
// C/D: col = l&15, row = (l>>4)*4 + reg, reg in [0,4).
typedef __attribute__((ext_vector_type(4))) float f32x4;

// Activation codes shared with python (smartcal_amd/ops/linear.py).
#define ACT_NONE 0
#define ACT_ELU 1
#define ACT_RELU 2
#define ACT_TANH 3

__device__ __forceinline__ float apply_act(float h, int act) {
  switch (act) {
    case ACT_ELU:  return h > 0.f ? h : __expf(h) - 1.f;
    case ACT_RELU: return h > 0.f ? h : 0.f;
    case ACT_TANH: return tanhf(h);
    default:       return h;
  }
}

// act'(h) expressed through the activation OUTPUT y (saves storing h):
// elu: y>0 ? 1 : y+1;  relu: y>0;  tanh: 1-y^2;  none: 1.
__device__ __forceinline__ float act_grad_from_y(float y, int act) {
  switch (act) {
    case ACT_ELU:  return y > 0.f ? 1.f : y + 1.f;
    case ACT_RELU: return y > 0.f ? 1.f : 0.f;
    case ACT_TANH: return 1.f - y * y;
    default:       return 1.f;
  }
}
