"""Demixing TD3 (CNN) training (reference `demixing_rl/main_td3.py`:
K−1 selection actions + PER)."""

import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import numpy as np

from smartcal_amd.envs.demix import DemixingEnv
from smartcal_amd.rl.td3_cnn import Agent
from smartcal_amd.utils.device import seed_everything
from _loop import run_training


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seed", default=0, type=int)
    ap.add_argument("--episodes", default=1000, type=int)
    ap.add_argument("--steps", default=7, type=int)
    ap.add_argument("--use_hint", action="store_true", default=False)
    ap.add_argument("--stations", default=62, type=int)
    args = ap.parse_args()
    seed_everything(args.seed)

    K = 6
    env = DemixingEnv(K=K, Nf=3, Ninf=128, Tdelta=10,
                      provide_hint=args.use_hint,
                      N_stations=args.stations, seed=args.seed)
    agent = Agent(gamma=0.99, batch_size=256, n_actions=K, tau=0.005,
                  max_mem_size=16000, input_dims=(1, 128, 128),
                  meta_dim=3 * K + 2, lr_a=3e-4, lr_c=3e-4, warmup=210,
                  prioritized=True, use_hint=args.use_hint)
    run_training(env, agent, args.episodes, args.steps,
                 provide_hint=args.use_hint, save_every=1)


if __name__ == "__main__":
    main()
