"""Elastic-net SAC hint vs no-hint learning curves (reference do.sh arms).

Incrementally dumps to gpurun_out/enet_hint_curves.json so partial
results survive a timeout.
"""
import json, sys
import numpy as np
import torch
sys.path.insert(0, ".")
from smartcal_amd.envs.enet import ENetEnv
from smartcal_amd.rl.sac import Agent
from smartcal_amd.utils.device import seed_everything

EPISODES = 150
STEPS = 10
out = {}
for arm, use_hint in (("hint", True), ("nohint", False)):
    out[arm] = {}
    for seed in (1, 2):
        seed_everything(seed)
        env = ENetEnv(20, 20, provide_hint=use_hint)
        agent = Agent(gamma=0.99, batch_size=64, n_actions=2, tau=0.005,
                      max_mem_size=1024, input_dims=[420], lr_a=1e-3,
                      lr_c=1e-3, reward_scale=20, alpha=0.03,
                      prioritized=use_hint, use_hint=use_hint)
        scores = []
        for ep in range(EPISODES):
            obs = env.reset(); done = False; sc = 0.0; n = 0
            while not done and n < STEPS:
                a = agent.choose_action(obs)
                if use_hint:
                    obs_, r, done, hint, info = env.step(a)
                    agent.store_transition(obs, a, r, obs_, done, hint)
                else:
                    obs_, r, done, info = env.step(a)
                    agent.store_transition(obs, a, r, obs_, done,
                                           np.zeros(2, np.float32))
                sc += float(r); agent.learn(); obs = obs_; n += 1
            scores.append(sc / n)
            if ep % 25 == 0:
                with open("gpurun_out/enet_hint_curves.json", "w") as f:
                    json.dump(out | {arm + "_partial": scores}, f)
        out[arm][seed] = scores
        print(f"{arm} seed {seed}: last25 mean "
              f"{np.mean(scores[-25:]):.3f}", flush=True)
with open("gpurun_out/enet_hint_curves.json", "w") as f:
    json.dump(out, f)
print("DONE", flush=True)
