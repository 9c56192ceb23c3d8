"""Elastic-net TD3 training (reference `elasticnet/main_td3.py`:
PER + hint, batch 64, mem 1024, warmup 100, noise 0.1, admm_rho=1)."""

import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

from smartcal_amd.envs.enet import ENetEnv
from smartcal_amd.rl.td3 import Agent
from smartcal_amd.utils.device import seed_everything
from _loop import run_training


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seed", default=0, type=int)
    ap.add_argument("--episodes", default=1000, type=int)
    ap.add_argument("--steps", default=4, type=int)
    ap.add_argument("--use_hint", action="store_true", default=True)
    args = ap.parse_args()
    seed_everything(args.seed)

    N = M = 20
    env = ENetEnv(M, N, provide_hint=args.use_hint)
    agent = Agent(gamma=0.99, batch_size=64, n_actions=2, tau=0.005,
                  max_mem_size=1024, input_dims=[N + N * M], lr_a=1e-3,
                  lr_c=1e-3, update_actor_interval=2, warmup=100,
                  noise=0.1, prioritized=True, use_hint=args.use_hint)
    agent.admm_rho = 1
    run_training(env, agent, args.episodes, args.steps,
                 provide_hint=args.use_hint)


if __name__ == "__main__":
    main()
