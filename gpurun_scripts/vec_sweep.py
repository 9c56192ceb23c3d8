"""GPU sweep: VecENetEnv batched-rollout throughput at E=1..4096.

VERDICT r1 item 5: measure the path that turns the ~1% GPU-busy
single-env RL loop into real utilization — E envs per enet_lbfgs_solve /
enet_influence launch. Reports pure env-step throughput and the full
train-loop (actor forward + env + batched store + learn) throughput.

Writes gpurun_out/vec_sweep.json.
"""

import json
import os
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import torch

from smartcal_amd.envs.vec_enet import VecENetEnv
from smartcal_amd.rl.sac import Agent
from smartcal_amd.utils.device import seed_everything

N = M = 20


def time_env_only(E, iters=50, warmup=10):
    env = VecENetEnv(E, M, N, device=torch.device("cuda"))
    env.reset()
    a = torch.rand(E, 2, device="cuda") * 2 - 1
    for _ in range(warmup):
        env.step(a)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        env.step(a)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    return E * iters / dt


def time_train_loop(E, iters=50, warmup=10):
    seed_everything(1)
    env = VecENetEnv(E, M, N, device=torch.device("cuda"))
    agent = Agent(gamma=0.99, batch_size=64, n_actions=2, tau=0.005,
                  max_mem_size=max(1024, E), input_dims=[N + N * M],
                  lr_a=1e-3, lr_c=1e-3, reward_scale=N, alpha=0.03)
    obs = env.reset()

    def one(obs):
        states = torch.cat((obs["eig"], obs["A"]), dim=1)
        with torch.no_grad():
            actions, _ = agent.actor.sample_normal(states,
                                                   reparameterize=False)
        obs_, rewards, done, _ = env.step(actions)
        states_ = torch.cat((obs_["eig"], obs_["A"]), dim=1)
        agent.replaymem.store_batch(states, actions, rewards, states_, done)
        agent.learn()
        return obs_

    for _ in range(warmup):
        obs = one(obs)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        obs = one(obs)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    return E * iters / dt


def main():
    out = {"env_only": {}, "train_loop": {}}
    for E in (1, 16, 64, 256, 1024, 4096):
        r = time_env_only(E)
        out["env_only"][E] = round(r, 1)
        print(f"env-only   E={E:5d}: {r:12.1f} env-steps/s", flush=True)
    for E in (16, 64, 256, 1024):
        r = time_train_loop(E)
        out["train_loop"][E] = round(r, 1)
        print(f"train-loop E={E:5d}: {r:12.1f} env-steps/s", flush=True)
    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/vec_sweep.json", "w") as f:
        json.dump(out, f, indent=1)


if __name__ == "__main__":
    main()
