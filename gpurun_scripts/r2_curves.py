"""Round-2 clean GPU learning curves — every arm.

Arms (reference do.sh / main_*.py semantics, figures/comparison.png):
  sac / td3(+PER) / ddpg           no-hint, 3 seeds
  sac_hint (PER + KLD-Lagrangian)  3 seeds
  td3_hint (PER + adaptive ADMM)   3 seeds
  td3_per_bf16                     bf16 compute parity arm, 2 seeds

1000 episodes x 5 steps each, degenerate-pair filter in (round-2 code).
Incremental dump to gpurun_out/r2_curves.json; NaN episodes counted.
"""

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
EPISODES = int(sys.argv[1]) if len(sys.argv) > 1 else 1000
STEPS = 5
OUT = Path("gpurun_out/r2_curves.json")
out = {"nan_episodes": {}}
if OUT.exists():
    out = json.loads(OUT.read_text())   # resume across calls


def make_agent(algo, use_hint, device):
    kw = dict(gamma=0.99, batch_size=64, n_actions=2, tau=0.005,
              max_mem_size=1024, input_dims=[N + N * M], lr_a=1e-3,
              lr_c=1e-3, device=device)
    if algo == "sac":
        from smartcal_amd.rl.sac import Agent
        return Agent(reward_scale=N, alpha=0.03, prioritized=use_hint,
                     use_hint=use_hint, **kw)
    if algo == "td3":
        from smartcal_amd.rl.td3 import Agent
        return Agent(update_actor_interval=2, warmup=100, noise=0.1,
                     prioritized=True, use_hint=use_hint, **kw)
    from smartcal_amd.rl.ddpg import Agent
    kw["tau"] = 0.001
    return Agent(**kw)


def train(algo, use_hint, seed, episodes=EPISODES):
    seed_everything(seed)
    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    env = ENetEnv(M, N, provide_hint=use_hint, device=device)
    agent = make_agent(algo, use_hint, device)
    scores = []
    nan_eps = 0
    t0 = time.time()
    for ep in range(episodes):
        obs = env.reset()
        total, n, done = 0.0, 0, False
        while not done and n < STEPS:
            a = agent.choose_action(obs)
            step_out = env.step(a)
            if use_hint:
                obs_, r, done, hint, _ = step_out
            else:
                obs_, r, done, _ = step_out
                hint = np.zeros(2, np.float32)
            agent.store_transition(obs, a, r, obs_, done, hint)
            agent.learn()
            total += float(r)
            obs = obs_
            n += 1
        s = total / max(n, 1)
        if not np.isfinite(s):
            nan_eps += 1
        scores.append(s)
    dt = time.time() - t0
    print(f"{algo}{'_hint' if use_hint else ''} seed {seed}: "
          f"{episodes} eps in {dt:.0f}s, last100 "
          f"{np.nanmean(scores[-100:]):.3f}, nan_eps {nan_eps}",
          flush=True)
    return scores, nan_eps


def run_arm(name, algo, use_hint, seeds, episodes=EPISODES):
    out.setdefault(name, {})
    for seed in seeds:
        if str(seed) in out[name]:
            continue
        scores, nan_eps = train(algo, use_hint, seed, episodes)
        out[name][str(seed)] = scores
        out["nan_episodes"][f"{name}:{seed}"] = nan_eps
        OUT.parent.mkdir(exist_ok=True)
        OUT.write_text(json.dumps(out))


def main():
    run_arm("sac", "sac", False, (1, 2, 3))
    run_arm("td3", "td3", False, (1, 2, 3))
    run_arm("ddpg", "ddpg", False, (1, 2, 3))
    run_arm("sac_hint", "sac", True, (1, 2, 3))
    run_arm("td3_hint", "td3", True, (1, 2, 3))
    from smartcal_amd.ops import linear as linear_ops
    linear_ops.set_compute_dtype("bf16")
    try:
        run_arm("td3_per_bf16", "td3", False, (1, 2))
    finally:
        linear_ops.set_compute_dtype("fp32")
    print("ALL ARMS DONE", flush=True)


if __name__ == "__main__":
    main()
