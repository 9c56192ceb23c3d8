// Empirical probe of the v_mfma_f32_16x16x32_bf16 A/B fragment lane
// mappings on gfx950 (C/D mapping is documented dtype-independent:
// col = lane&15, row = (lane>>4)*4 + reg).
//
// Method: zero both fragments except element r0 of lane l0 set to 1 in A
// (resp. B); with B[k][j] = k (resp. A[i][k] = k) the product D exposes
// (i, k) (resp. (j, k)) of that fragment slot. Integers 0..31 are exact
// in bf16. Output: OUTA/OUTB[l0*8+r0] = i*100 + k (j*100 + k).
//
// Build on-box:  hipcc --offload-arch=gfx950 -o probe mfma_bf16_probe.hip
#include <hip/hip_runtime.h>
#include <cstdio>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

__global__ void probe_kernel(int* OUTA, int* OUTB) {
  const int lane = threadIdx.x;  // one wave
  for (int l0 = 0; l0 < 64; ++l0) {
    for (int r0 = 0; r0 < 8; ++r0) {
      // ---- A probe ----
      bf16x8 a = {0, 0, 0, 0, 0, 0, 0, 0};
      if (lane == l0) a[r0] = (__bf16)1.0f;
      // B[k][j] = k for all j: every lane's B slot value = its k.
      // We don't know B's layout yet, but "value = k" only needs k —
      // bootstrap: B's k is ALSO unknown. Break the circularity by
      // probing with B = broadcast over unknown j: set every B element
      // of every lane to the candidate. Instead: first find A's (i)
      // only, with B all-ones (D row = i, D value = 1 summed over one k).
      bf16x8 ones;
      for (int r = 0; r < 8; ++r) ones[r] = (__bf16)1.0f;
      f32x4 acc = {0.f, 0.f, 0.f, 0.f};
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, ones, acc, 0, 0, 0);
      // find nonzero entry of D across the wave: each lane holds
      // D[(lane>>4)*4+r][lane&15]
      int found_i = -1;
      for (int r = 0; r < 4; ++r)
        if (acc[r] != 0.f) found_i = (lane >> 4) * 4 + r;
      // reduce across wave (max; exactly one row nonzero, all 16 cols)
      for (int off = 32; off > 0; off >>= 1)
        found_i = max(found_i, __shfl_xor(found_i, off, 64));

      // now k: need B[k][j] = k. We must know B's layout... use the
      // transpose trick: compute with A' = probe as before but B' such
      // that B'[k][j] = k via a SECOND mfma once B's layout is known.
      // Instead probe k directly: run 32 trials with B one-hot in k:
      // B_t[k][j] = (k == t). t-th trial nonzero => k = t. B one-hot in
      // k means: every lane sets the slots whose k == t... unknown again.
      // FINAL approach that needs no B knowledge: A-k probe via
      // D = A_kval * B_probe in the B section below (symmetric).
      // Here instead: exploit that sum_j D[i][j] over the found row
      // equals 16 regardless of k (B all ones) — so k is NOT
      // recoverable from this trial. Use the B section's result.
      if (lane == 0) OUTA[l0 * 8 + r0] = found_i;

      // ---- B probe: one-hot B, A[i][k] known per A's OWN slot ----
      // A[i][k] = k needs A's k-map, which we get iteratively: assume
      // candidate map k = (l>>4)*4 + (r&3) + (r>>2)*16 ("two halves").
      // The self-consistency check in python validates both maps
      // against torch matmul on random data; if it fails, candidate 2
      // (k = (l>>4)*8 + r) is validated instead (PROBE_ALT build).
      bf16x8 akv;
      for (int r = 0; r < 8; ++r) {
#ifdef ALT_LAYOUT
        int k = (lane >> 4) * 8 + r;
#else
        int k = (lane >> 4) * 4 + (r & 3) + (r >> 2) * 16;
#endif
        akv[r] = (__bf16)(float)k;
      }
      bf16x8 b = {0, 0, 0, 0, 0, 0, 0, 0};
      if (lane == l0) b[r0] = (__bf16)1.0f;
      f32x4 acc2 = {0.f, 0.f, 0.f, 0.f};
      acc2 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(akv, b, acc2, 0, 0, 0);
      // D[i][j0] = k0 for every i (A col k0 = const k0 across i);
      // lane&15 == j0 lanes hold it.
      int found_j = -1, found_k = -1;
      for (int r = 0; r < 4; ++r)
        if (acc2[r] != 0.f) { found_j = lane & 15; found_k = (int)acc2[r]; }
      for (int off = 32; off > 0; off >>= 1) {
        found_j = max(found_j, __shfl_xor(found_j, off, 64));
        found_k = max(found_k, __shfl_xor(found_k, off, 64));
      }
      if (lane == 0) OUTB[l0 * 8 + r0] = found_j * 100 + found_k;
    }
  }
}

// Random-data GEMM check of the assumed A/B maps: 16x16x32 single tile.
__global__ void gemm_check_kernel(const float* A, const float* B, float* D) {
  const int lane = threadIdx.x;
  bf16x8 af, bf;
  for (int r = 0; r < 8; ++r) {
#ifdef ALT_LAYOUT
    int k = (lane >> 4) * 8 + r;
#else
    int k = (lane >> 4) * 4 + (r & 3) + (r >> 2) * 16;
#endif
    af[r] = (__bf16)A[(lane & 15) * 32 + k];
    bf[r] = (__bf16)B[k * 16 + (lane & 15)];
  }
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc, 0, 0, 0);
  for (int r = 0; r < 4; ++r)
    D[((lane >> 4) * 4 + r) * 16 + (lane & 15)] = acc[r];
}

int main() {
  int *da, *db;
  hipMalloc(&da, 512 * sizeof(int));
  hipMalloc(&db, 512 * sizeof(int));
  hipLaunchKernelGGL(probe_kernel, dim3(1), dim3(64), 0, 0, da, db);
  int ha[512], hb[512];
  hipMemcpy(ha, da, sizeof(ha), hipMemcpyDeviceToHost);
  hipMemcpy(hb, db, sizeof(hb), hipMemcpyDeviceToHost);
  printf("A map (lane,reg)->i:\n");
  for (int l = 0; l < 64; l += 16)
    for (int r = 0; r < 8; ++r)
      printf("l=%d r=%d i=%d\n", l, r, ha[l * 8 + r]);
  printf("B map (lane,reg)->j*100+k:\n");
  for (int l = 0; l < 64; l += 16)
    for (int r = 0; r < 8; ++r)
      printf("l=%d r=%d jk=%d\n", l, r, hb[l * 8 + r]);

  // random GEMM check vs host
  float hA[16 * 32], hB[32 * 16], hD[256], ref[256];
  unsigned s = 12345;
  auto rnd = [&]() {
    s = s * 1664525u + 1013904223u;
    return ((s >> 8) & 0xFFFF) / 65536.0f - 0.5f;
  };
  for (int i = 0; i < 16 * 32; ++i) hA[i] = rnd();
  for (int i = 0; i < 32 * 16; ++i) hB[i] = rnd();
  // bf16-truncate host copies for the reference
  auto tobf = [](float x) {
    unsigned u;
    __builtin_memcpy(&u, &x, 4);
    u = (u + 0x8000) & 0xFFFF0000u;  // round-to-nearest-even approx
    float y;
    __builtin_memcpy(&y, &u, 4);
    return y;
  };
  for (int i = 0; i < 16; ++i)
    for (int j = 0; j < 16; ++j) {
      float acc = 0.f;
      for (int k = 0; k < 32; ++k)
        acc += tobf(hA[i * 32 + k]) * tobf(hB[k * 16 + j]);
      ref[i * 16 + j] = acc;
    }
  float *dA, *dB, *dD;
  hipMalloc(&dA, sizeof(hA));
  hipMalloc(&dB, sizeof(hB));
  hipMalloc(&dD, sizeof(hD));
  hipMemcpy(dA, hA, sizeof(hA), hipMemcpyHostToDevice);
  hipMemcpy(dB, hB, sizeof(hB), hipMemcpyHostToDevice);
  hipLaunchKernelGGL(gemm_check_kernel, dim3(1), dim3(64), 0, 0, dA, dB, dD);
  hipMemcpy(hD, dD, sizeof(hD), hipMemcpyDeviceToHost);
  float maxerr = 0.f;
  for (int i = 0; i < 256; ++i) {
    float e = fabsf(hD[i] - ref[i]);
    if (e > maxerr) maxerr = e;
  }
  printf("gemm check maxerr = %g  (%s)\n", maxerr,
         maxerr < 2e-2 ? "PASS" : "FAIL");
  return maxerr < 2e-2 ? 0 : 1;
}
