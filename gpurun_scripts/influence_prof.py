"""Influence-core timing breakdown on MI355X.

Times hessianres / dsolutions_r / dresiduals at LOFAR scale (N=62,
B=1891) before/after the cgemm kernel, plus whole CalibEnv / DemixingEnv
steps. Writes gpurun_out/influence_prof.json.
"""

import json
import os
import sys
import time
from pathlib import Path

import numpy as np
import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

from smartcal_amd.radio import hessian as hs

DEV = torch.device("cuda")


def timeit(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3  # ms


def main():
    out = {}
    rng = np.random.default_rng(0)
    N, K, Tdelta = 62, 6, 10
    B = N * (N - 1) // 2
    S = B * Tdelta
    C = torch.from_numpy((rng.standard_normal((K, S, 4))
                          + 1j * rng.standard_normal((K, S, 4))
                          ).astype(np.complex64)).to(DEV) * 0.1
    J = torch.from_numpy((rng.standard_normal((K, 2 * N, 2))
                          + 1j * rng.standard_normal((K, 2 * N, 2))
                          ).astype(np.complex64)).to(DEV)
    R = torch.from_numpy((rng.standard_normal((2 * S, 2))
                          + 1j * rng.standard_normal((2 * S, 2))
                          ).astype(np.complex64)).to(DEV)

    H = hs.hessianres(R, C, J, N)
    dJ = hs.dsolutions_r(C, J, N, H)
    out["hessianres_ms"] = round(timeit(lambda: hs.hessianres(R, C, J, N)), 2)
    out["dsolutions_ms"] = round(
        timeit(lambda: hs.dsolutions_r(C, J, N, H)), 2)
    out["dresiduals_r_ms"] = round(
        timeit(lambda: hs.dresiduals_r(C, J, N, dJ, False)), 2)
    out["dresiduals_rk_ms"] = round(
        timeit(lambda: hs.dresiduals_rk(C, J, N, dJ, False)), 2)
    out["dres_colmeans_ms"] = round(
        timeit(lambda: hs.dres_colmeans(C, J, N, H)), 2)
    out["dres_colmeans_perk_ms"] = round(
        timeit(lambda: hs.dres_colmeans(C, J, N, H, per_k=True)), 2)
    # the dsolutions GEMM alone, old path vs kernel
    eye = torch.eye(4 * N, dtype=C.dtype, device=DEV)
    Ainv = torch.linalg.inv(H + 1e-12 * eye)
    AdV = (torch.randn(8, K, 4 * N, B) + 1j * torch.randn(8, K, 4 * N, B)) \
        .to(torch.complex64).to(DEV)
    Bv = AdV.permute(1, 0, 2, 3).contiguous().reshape(K * 8, 4 * N, B)
    from smartcal_amd import ops
    out["cgemm_kernel_ms"] = round(
        timeit(lambda: ops.ext().cgemm_nn_bcast(Ainv.contiguous(), Bv, 8)),
        2)
    out["rocblas_gemm_ms"] = round(
        timeit(lambda: Ainv.unsqueeze(1) @ AdV.permute(1, 0, 2, 3)), 2)

    # whole env steps
    from smartcal_amd.envs.calib import CalibEnv
    from smartcal_amd.envs.demix import DemixingEnv
    env = CalibEnv(M=10, N_stations=62, Nf=8, Ts=2, Tdelta=10,
                   device=DEV, seed=1)
    env.reset()
    a = np.random.rand(2 * 10) * 2 - 1
    t0 = time.perf_counter()
    env.step(a)
    torch.cuda.synchronize()
    out["calib_step_s"] = round(time.perf_counter() - t0, 3)
    denv = DemixingEnv(K=6, Nf=3, Ninf=128, Tdelta=10, Ts=2,
                       provide_influence=True, N_stations=62, device=DEV,
                       seed=1)
    denv.reset()
    da = np.random.rand(6) * 2 - 1
    t0 = time.perf_counter()
    denv.step(da)
    torch.cuda.synchronize()
    out["demix_step_s"] = round(time.perf_counter() - t0, 3)

    print(json.dumps(out, indent=1))
    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/influence_prof.json", "w") as f:
        json.dump(out, f, indent=1)


if __name__ == "__main__":
    main()
