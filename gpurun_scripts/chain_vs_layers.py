"""Chain kernel vs per-layer fused calls for the SAC step's MLP shapes."""
import sys, time, torch
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))
import smartcal_amd.ops as ops

DEV = torch.device("cuda")

def timeit(fn, iters=300):
    for _ in range(30):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6

for name, B, dims in (("actor3", 64, [420, 512, 256, 128]),
                      ("critS2", 64, [422, 512, 256]),
                      ("critA2", 64, [2, 128, 64]),
                      ("act1",   1, [420, 512, 256, 128])):
    L = len(dims) - 1
    x = torch.randn(B, dims[0], device=DEV)
    Ws = [torch.randn(dims[i+1], dims[i], device=DEV) / dims[i] ** 0.5
          for i in range(L)]
    bs = [torch.randn(dims[i+1], device=DEV) for i in range(L)]
    gs = [torch.rand(dims[i+1], device=DEV) + 0.5 for i in range(L)]
    bes = [torch.randn(dims[i+1], device=DEV) for i in range(L)]
    acts = [1] * L
    us_chain = timeit(lambda: ops.ext().mlp_chain_fwd(
        x, Ws, bs, gs, bes, acts, False))
    def per_layer():
        h = x
        for i in range(L):
            h, _, _ = ops.ext().fused_linear_fwd(h.contiguous(), Ws[i],
                                                 bs[i], gs[i], bes[i], 1,
                                                 True)
        return h
    us_layers = timeit(per_layer)
    print(f"{name:8s} B={B:3d} L={L}: chain {us_chain:7.1f} us  "
          f"per-layer {us_layers:7.1f} us", flush=True)
