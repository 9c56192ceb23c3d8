"""Microbenchmark the enet solver/influence kernels (GPU box only)."""
import time
import sys, pathlib
sys.path.insert(0, str(pathlib.Path(__file__).resolve().parents[1]))

import torch

import smartcal_amd.ops as ops

DEV = torch.device("cuda:0")
torch.manual_seed(0)
N = M = 20


def timeit(fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


for E in [1, 8, 64, 256]:
    A = torch.randn(E, N, M, device=DEV)
    A /= A.reshape(E, -1).norm(dim=1).reshape(E, 1, 1)
    y = torch.randn(E, N, device=DEV) * 0.3
    rho = torch.full((E, 2), 0.05, device=DEV)
    pen = torch.zeros(E, device=DEV)
    for ep in ([1, 5, 20] if E == 1 else [20]):
        us = timeit(lambda: ops.ext().enet_lbfgs_solve(A, y, rho, ep, 10, 7))
        print(f"E={E:4d} epochs={ep:2d}: solve {us:9.1f} us "
              f"({us/E:7.2f} us/env)")
    x, Y, S, nh = ops.ext().enet_lbfgs_solve(A, y, rho, 20, 10, 7)
    us = timeit(lambda: ops.ext().enet_influence(A, y, x, Y, S, nh, pen, rho))
    print(f"E={E:4d} influence: {us:9.1f} us ({us/E:7.2f} us/env)  "
          f"nhist={int(nh[0])}")
