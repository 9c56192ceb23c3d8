"""Localize the N=96 launch failure (run with AMD_SERIALIZE_KERNEL=3)."""
import sys, numpy as np, torch
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))
from smartcal_amd.radio import hessian as hs

def stage(name, fn):
    try:
        fn()
        torch.cuda.synchronize()
        print(f"{name}: OK", flush=True)
    except Exception as e:
        print(f"{name}: FAIL {type(e).__name__}: {e}", flush=True)
        raise SystemExit(1)

rng = np.random.default_rng(0)
for N in (62, 96):
    K, T = 3, 10
    B = N * (N - 1) // 2
    S = B * T
    C = torch.from_numpy((rng.standard_normal((K, S, 4))
                          + 1j * rng.standard_normal((K, S, 4))
                          ).astype(np.complex64)).cuda() * 0.1
    J = torch.from_numpy((rng.standard_normal((K, 2 * N, 2))
                          + 1j * rng.standard_normal((K, 2 * N, 2))
                          ).astype(np.complex64)).cuda()
    R = torch.from_numpy((rng.standard_normal((2 * S, 2))
                          + 1j * rng.standard_normal((2 * S, 2))
                          ).astype(np.complex64)).cuda()
    H = None
    def _h():
        global H
        H = hs.hessianres(R, C, J, N)
    stage(f"hessianres N={N}", _h)
    stage(f"colmeans N={N}", lambda: hs.dres_colmeans(C, J, N, H))
    stage(f"colmeans_perk N={N}",
          lambda: hs.dres_colmeans(C, J, N, H, per_k=True))
# now the full env path
from smartcal_amd.envs.calib import CalibEnv
env = CalibEnv(M=10, N_stations=96, Nf=2, Ts=1, Tdelta=5,
               device=torch.device("cuda"), seed=1)
stage("CalibEnv N=96 reset", env.reset)
print("ALL OK", flush=True)
