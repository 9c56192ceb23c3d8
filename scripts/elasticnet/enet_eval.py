"""Evaluate a trained elastic-net agent against classic GridSearchCV.

Parity with `elasticnet/enet_eval.py:85-112`: on the same noisy problem
instance, compare the RL agent's (λ1, λ2) against sklearn grid search,
reporting the relative solution errors of both.
"""

import argparse
import sys
from pathlib import Path

import numpy as np
import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))

from smartcal_amd.envs.enet import ENetEnv
from smartcal_amd.rl.sac import Agent


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--episodes", default=10, type=int)
    ap.add_argument("--seed", default=0, type=int)
    args = ap.parse_args()
    np.random.seed(args.seed)
    torch.manual_seed(args.seed)

    N = M = 20
    env = ENetEnv(M, N, provide_hint=True)
    agent = Agent(gamma=0.99, batch_size=64, n_actions=2, tau=0.005,
                  max_mem_size=1024, input_dims=[N + N * M], lr_a=1e-3,
                  lr_c=1e-3, reward_scale=N, alpha=0.03)
    agent.load_models_for_eval()

    for ep in range(args.episodes):
        obs = env.reset()
        action = agent.choose_action(obs)
        # RL-chosen regularization, keeping the same noise realization
        _, reward_rl, _, _, _ = env.step(action, keepnoise=True)
        err_rl = env.solution_error()
        # classic hint (grid search) on the same instance
        hint = env.get_hint()
        _, reward_gs, _, _, _ = env.step(hint, keepnoise=True)
        err_gs = env.solution_error()
        print(f"episode {ep}: RL err {err_rl:.4f} (r={float(reward_rl):.3f})"
              f"  GridSearch err {err_gs:.4f} (r={float(reward_gs):.3f})")


if __name__ == "__main__":
    main()
