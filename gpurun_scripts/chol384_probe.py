import torch, sys
def stage(name, fn):
    try:
        r = fn()
        torch.cuda.synchronize()
        print(f"{name}: OK", flush=True)
        return True
    except Exception as e:
        print(f"{name}: FAIL {type(e).__name__}: {str(e)[:90]}", flush=True)
        torch.cuda.synchronize()
        return False
for n in (248, 384, 512):
    for K in (1, 3):
        H = torch.randn(K, n, n, dtype=torch.complex64, device="cuda")
        H = H @ H.mH + 0.5 * torch.eye(n, dtype=torch.complex64,
                                       device="cuda")
        W = torch.randn(K, 4, n, dtype=torch.complex64, device="cuda")
        stage(f"cholesky_ex n={n} K={K}",
              lambda: torch.linalg.cholesky_ex(H))
        stage(f"lu solve    n={n} K={K}",
              lambda: torch.linalg.solve(H.mT, W.mT))
        stage(f"inv         n={n} K={K}", lambda: torch.linalg.inv(H))
