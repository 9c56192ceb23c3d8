"""Fuzzy-demixing SAC training.

Reference `demixing_fuzzy/main_sac.py:14-115`: action = 24(K−1)+8 fuzzy
membership parameters, --use_influence gates the CNN branch, reward
shaping ×10 if > 0.01 floored at −10.
"""

import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

from smartcal_amd.envs.demix_fuzzy import FuzzyDemixingEnv
from smartcal_amd.rl.sac_cnn import Agent
from smartcal_amd.utils.device import seed_everything
from _loop import run_training


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seed", default=0, type=int)
    ap.add_argument("--episodes", default=1000, type=int)
    ap.add_argument("--steps", default=7, type=int)
    ap.add_argument("--use_hint", action="store_true", default=False)
    ap.add_argument("--use_influence", action="store_true", default=False)
    ap.add_argument("--memory", default=16000, type=int)
    ap.add_argument("--batch_size", default=256, type=int)
    ap.add_argument("--stations", default=62, type=int)
    args = ap.parse_args()
    seed_everything(args.seed)

    K = 6
    env = FuzzyDemixingEnv(K=K, Nf=3, Ninf=128, Tdelta=10,
                           provide_hint=args.use_hint,
                           provide_influence=args.use_influence,
                           N_stations=args.stations, seed=args.seed)
    n_actions = 24 * (K - 1) + 8
    agent = Agent(gamma=0.99, batch_size=args.batch_size,
                  n_actions=n_actions, tau=0.005,
                  max_mem_size=args.memory, input_dims=(1, 128, 128),
                  meta_dim=5 * K + 2, lr_a=3e-4, lr_c=3e-4,
                  use_hint=args.use_hint,
                  use_influence=args.use_influence)

    def shape(r):
        r = r * 10 if r > 0.01 else r
        return max(r, -10.0)

    run_training(env, agent, args.episodes, args.steps,
                 provide_hint=args.use_hint, reward_shaping=shape,
                 save_every=1)


if __name__ == "__main__":
    main()
