import time, sys, pathlib
sys.path.insert(0, str(pathlib.Path(__file__).resolve().parents[1]))
import torch
import smartcal_amd.ops as ops
DEV = torch.device("cuda:0")
def timeit(fn, iters=200):
    for _ in range(20): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter()-t0)/iters*1e6
for (B,K,N) in [(1,420,512),(64,420,512),(64,512,256),(64,256,128),(64,128,4),(64,320,1),(256,420,512)]:
    x = torch.randn(B,K,device=DEV); W = torch.randn(N,K,device=DEV)*0.05
    b = torch.randn(N,device=DEV); g = torch.rand(N,device=DEV)+0.5; be = torch.randn(N,device=DEV)
    us = timeit(lambda: ops.ext().fused_linear_fwd(x,W,b,g,be,1,True))
    print(f"fwd B={B:4d} K={K:4d} N={N:4d}: {us:7.2f} us")
dz = torch.randn(64,512,device=DEV); Wm = torch.randn(512,420,device=DEV); xx = torch.randn(64,420,device=DEV)
print("gemm_nn 64x512@512x420:", round(timeit(lambda: ops.ext().mfma_gemm_nn(dz,Wm)),2), "us")
print("gemm_tn+bias:", round(timeit(lambda: ops.ext().mfma_gemm_tn_bias(dz,xx)),2), "us")
