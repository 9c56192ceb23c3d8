"""Calibration-tuning TD3 (CNN) training (reference
`calibration/main_td3.py`)."""

import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

from smartcal_amd.envs.calib import CalibEnv
from smartcal_amd.rl.td3_cnn import Agent
from smartcal_amd.utils.device import seed_everything
from _loop import run_training


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seed", default=0, type=int)
    ap.add_argument("--episodes", default=50, type=int)
    ap.add_argument("--steps", default=4, type=int)
    ap.add_argument("--use_hint", action="store_true", default=False)
    ap.add_argument("--M", default=10, type=int)
    ap.add_argument("--stations", default=62, type=int)
    ap.add_argument("--arch", default="cnn", choices=("cnn", "transformer"))
    args = ap.parse_args()
    seed_everything(args.seed)

    M = args.M
    env = CalibEnv(M=M, provide_hint=args.use_hint,
                   N_stations=args.stations, seed=args.seed)
    agent = Agent(arch=args.arch,
                  gamma=0.99, batch_size=32, n_actions=2 * M, tau=0.005,
                  max_mem_size=10000, input_dims=(1, 128, 128), M=M,
                  lr_a=1e-3, lr_c=1e-3, warmup=100, noise=0.1,
                  prioritized=True, use_hint=args.use_hint)
    run_training(env, agent, args.episodes, args.steps,
                 provide_hint=args.use_hint)


if __name__ == "__main__":
    main()
