"""Vectorized-rollout SAC training: E elastic-net envs per solver launch.

Beyond the reference (whose env is strictly single-instance): every
iteration steps E independent problems with ONE batched in-kernel
L-BFGS + influence launch (`VecENetEnv`), stores E transitions and runs
one learn step. On an MI355X this turns the rollout from E sequential
2 ms env steps into one ~2 ms batched launch — the natural way to fill
the replay at device speed. The agent and its checkpoint layout are the
standard SAC ones.
"""

import argparse
import pickle
import sys
from pathlib import Path

import numpy as np
import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))

from smartcal_amd.envs.vec_enet import VecENetEnv
from smartcal_amd.rl.sac import Agent
from smartcal_amd.utils.device import seed_everything

N = M = 20


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seed", default=0, type=int)
    ap.add_argument("--envs", default=16, type=int)
    ap.add_argument("--iters", default=300, type=int,
                    help="batched env steps (each = E transitions)")
    ap.add_argument("--reset-every", default=5, type=int)
    args = ap.parse_args()
    seed_everything(args.seed)

    env = VecENetEnv(args.envs, M, N)
    agent = Agent(gamma=0.99, batch_size=64, n_actions=2, tau=0.005,
                  max_mem_size=1024, input_dims=[N + N * M], lr_a=1e-3,
                  lr_c=1e-3, reward_scale=N, alpha=0.03)
    dev = agent.device

    obs = env.reset()
    zeros2 = np.zeros(2, dtype=np.float32)
    scores = []
    for it in range(args.iters):
        states = torch.cat((obs["eig"], obs["A"]), dim=1).to(dev)
        with torch.no_grad():
            actions, _ = agent.actor.sample_normal(states,
                                                   reparameterize=False)
        obs_, rewards, done, _ = env.step(actions.to(env.device))
        states_ = torch.cat((obs_["eig"], obs_["A"]), dim=1)
        # one batched device-resident write for all E transitions
        agent.replaymem.store_batch(states, actions,
                                    torch.as_tensor(rewards), states_, done)
        agent.learn()
        scores.append(float(rewards.mean()))
        if (it + 1) % args.reset_every == 0:
            obs = env.reset()
        else:
            obs = obs_
        if it % 20 == 0:
            print(f"iter {it} mean reward {np.mean(scores[-20:]):.3f} "
                  f"({args.envs} envs/step)")
    agent.save_models()
    with open("scores.pkl", "wb") as f:
        pickle.dump(scores, f)


if __name__ == "__main__":
    main()
