"""Reproduce the reference's elastic-net reward-vs-episode comparison
(figures/comparison.png: SAC and TD3 converge high, DDPG lower) on
MI355X, writing curves JSON + PNG under gpurun_out/."""

import json
import sys
import time
from pathlib import Path

import numpy as np
import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

from smartcal_amd.envs.enet import ENetEnv
from smartcal_amd.utils.device import seed_everything

N = M = 20
EPISODES = int(sys.argv[1]) if len(sys.argv) > 1 else 400
SEED = int(sys.argv[2]) if len(sys.argv) > 2 else 1
STEPS = 5


def train(algo: str, seed: int):
    seed_everything(seed)
    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    env = ENetEnv(M, N, provide_hint=False, device=device)
    kw = dict(gamma=0.99, batch_size=64, n_actions=2, tau=0.005,
              max_mem_size=1024, input_dims=[N + N * M], lr_a=1e-3,
              lr_c=1e-3, device=device)
    if algo == "sac":
        from smartcal_amd.rl.sac import Agent
        agent = Agent(reward_scale=N, alpha=0.03, **kw)
    elif algo == "td3":
        from smartcal_amd.rl.td3 import Agent
        # reference main_td3.py default: prioritized replay on
        agent = Agent(update_actor_interval=2, warmup=100, noise=0.1,
                      prioritized=True, **kw)
    else:
        from smartcal_amd.rl.ddpg import Agent
        kw["tau"] = 0.001
        agent = Agent(**kw)
    scores = []
    t0 = time.time()
    for ep in range(EPISODES):
        obs = env.reset()
        total = 0.0
        for _ in range(STEPS):
            a = agent.choose_action(obs)
            obs2, r, done, info = env.step(a)
            agent.store_transition(obs, a, r, obs2, done,
                                   np.zeros(2, np.float32))
            agent.learn()
            total += float(r)
            obs = obs2
        scores.append(total / STEPS)
    print(f"{algo}: {EPISODES} episodes in {time.time() - t0:.1f}s, "
          f"final avg100 {np.mean(scores[-100:]):.3f}")
    return scores


def main():
    seeds = (1, 2, 3)
    out = {}
    for algo in ("sac", "td3", "ddpg"):
        out[algo] = {str(s): train(algo, seed=s) for s in seeds}
    Path("gpurun_out").mkdir(exist_ok=True)
    with open("gpurun_out/learning_curves.json", "w") as f:
        json.dump(out, f)
    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt
    plt.figure(figsize=(7, 4.5))
    colors = {"sac": "C0", "td3": "C1", "ddpg": "C2"}
    for algo, runs in out.items():
        sm = np.stack([np.convolve(np.asarray(s), np.ones(25) / 25,
                                   mode="valid") for s in runs.values()])
        med = np.median(sm, axis=0)
        plt.plot(med, color=colors[algo], label=algo.upper())
        plt.fill_between(np.arange(sm.shape[1]), sm.min(0), sm.max(0),
                         color=colors[algo], alpha=0.15)
    plt.xlabel("episode")
    plt.ylabel("score (25-episode moving average)")
    plt.title(f"Elastic-net tuning on 1x MI355X "
              f"(N=M={N}, batch 64; median of {len(seeds)} seeds)")
    plt.legend()
    plt.grid(alpha=0.3)
    plt.tight_layout()
    plt.savefig("gpurun_out/learning_curves.png", dpi=130)


if __name__ == "__main__":
    main()
