"""Demixing SAC hint vs no-hint WITH influence maps on (CNN agents).

Round-1 curves ran metadata-only (provide_influence=False); after the
round-2 influence-core work (0.019 s/step) the full CNN config is
curve-feasible. Time-budgeted: runs as many episodes as fit.

Usage: python demix_influence_curves.py [episodes] [budget_s_per_arm]
"""

import json
import sys
import time
from pathlib import Path

import numpy as np
import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

from smartcal_amd.envs.demix import DemixingEnv
from smartcal_amd.rl.sac_cnn import Agent
from smartcal_amd.utils.device import seed_everything

EPISODES = int(sys.argv[1]) if len(sys.argv) > 1 else 300
BUDGET_S = float(sys.argv[2]) if len(sys.argv) > 2 else 420.0
K = 6
STEPS = 7
OUT = Path("gpurun_out/demix_influence_curves.json")
out = json.loads(OUT.read_text()) if OUT.exists() else {}


def train(use_hint: bool, seed: int = 5):
    seed_everything(seed)
    env = DemixingEnv(K=K, Nf=3, Ninf=128, Tdelta=10, Ts=2,
                      provide_hint=use_hint, provide_influence=True,
                      N_stations=62, device="cuda", seed=seed)
    agent = Agent(gamma=0.99, batch_size=64, n_actions=K, tau=0.005,
                  max_mem_size=4000, input_dims=(1, 128, 128),
                  meta_dim=3 * K + 2, lr_a=3e-4, lr_c=3e-4,
                  use_hint=use_hint, use_influence=True,
                  hint_threshold=0.0, admm_rho=1.0,
                  device=torch.device("cuda"))
    scores = []
    t0 = time.time()
    for ep in range(EPISODES):
        obs = env.reset()
        hint = np.zeros(K, np.float32)
        tot, n, done = 0.0, 0, False
        while not done and n < STEPS:
            a = agent.choose_action(obs)
            step_out = env.step(a)
            if use_hint:
                obs_, r, done, hint, _ = step_out
            else:
                obs_, r, done, _ = step_out
            r_shaped = float(r) * 10 if r > 0 else float(r)
            agent.store_transition(obs, a, r_shaped, obs_, done, hint)
            agent.learn()
            tot += float(r)
            obs = obs_
            n += 1
        scores.append(tot / max(n, 1))
        if time.time() - t0 > BUDGET_S:
            break
        if ep % 20 == 0:
            OUT.write_text(json.dumps(
                out | {("hint" if use_hint else "nohint")
                       + "_partial": scores}))
    dt = time.time() - t0
    print(f"{'hint' if use_hint else 'nohint'}: {len(scores)} eps in "
          f"{dt:.0f}s, first50 {np.mean(scores[:50]):.4f} last50 "
          f"{np.mean(scores[-50:]):.4f}", flush=True)
    return scores


def main():
    for arm, use_hint in (("nohint", False), ("hint", True)):
        if arm in out:
            continue
        out[arm] = train(use_hint)
        out.pop(arm + "_partial", None)
        OUT.write_text(json.dumps(out))
    print("DONE", flush=True)


if __name__ == "__main__":
    main()
