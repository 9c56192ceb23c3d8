"""Reproduce the reference's demixing result (figures/calibration_rewards
.png: hint-trained SAC's reward rises, no-hint stays flat) on MI355X at
full LOFAR scale (N=62)."""

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

EPISODES = int(sys.argv[1]) if len(sys.argv) > 1 else 60
STEPS = 7
K = 6


def train(use_hint: bool, seed: int = 5):
    seed_everything(seed)
    env = DemixingEnv(K=K, Nf=3, Ninf=128, Tdelta=10, Ts=2,
                      provide_hint=use_hint, provide_influence=False,
                      N_stations=62, device="cuda", seed=seed)
    agent = Agent(gamma=0.99, batch_size=64, n_actions=K, tau=0.005,
                  max_mem_size=4000, input_dims=(1, 128, 128),
                  meta_dim=3 * K + 2, lr_a=3e-4, lr_c=3e-4,
                  use_hint=use_hint, use_influence=False,
                  hint_threshold=0.0, admm_rho=1.0,
                  device=torch.device("cuda"))
    scores = []
    t0 = time.time()
    for ep in range(EPISODES):
        obs = env.reset()
        tot = 0.0
        hint = np.zeros(K, np.float32)
        for _ in range(STEPS):
            a = agent.choose_action(obs) if ep >= 3 \
                else env.action_space.sample().reshape(-1)
            out = env.step(a)
            if use_hint:
                obs2, r, done, hint, _ = out
            else:
                obs2, r, done, _ = out
            r10 = r * 10 if r > 0 else r
            agent.store_transition(obs, a, r10, obs2, done, hint)
            agent.learn()
            tot += float(r)
            obs = obs2
        scores.append(tot / STEPS)
        if ep % 10 == 9:
            print(f"hint={use_hint} ep{ep}: avg10="
                  f"{np.mean(scores[-10:]):.4f} ({time.time() - t0:.0f}s)")
    return scores


def main():
    out = {"hint": train(True), "nohint": train(False)}
    Path("gpurun_out").mkdir(exist_ok=True)
    with open("gpurun_out/demix_curves.json", "w") as f:
        json.dump(out, f)
    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt
    plt.figure(figsize=(7, 4.5))
    for name, s in out.items():
        sm = np.convolve(np.asarray(s), np.ones(10) / 10, mode="valid")
        plt.plot(sm, label=name)
    plt.xlabel("episode")
    plt.ylabel("score (10-episode moving average)")
    plt.title(f"Demixing SAC on 1x MI355X (N=62, K={K}; hint vs no hint)")
    plt.legend()
    plt.grid(alpha=0.3)
    plt.tight_layout()
    plt.savefig("gpurun_out/demix_curves.png", dpi=130)


if __name__ == "__main__":
    main()
