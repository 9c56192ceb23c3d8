"""Collect (metadata, hint) pairs from the demixing env for distillation.

Reference `demixing_rl/makedata.py:27-35`: run episodes, store the env
metadata and the exhaustive-sweep hint into a TrainingBuffer.
"""

import argparse
import sys
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))

from smartcal_amd.envs.demix import DemixingEnv
from smartcal_amd.models import TrainingBuffer


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--samples", default=100, type=int)
    ap.add_argument("--stations", default=26, type=int)
    ap.add_argument("--seed", default=0, type=int)
    ap.add_argument("--out", default="databuffer.npy")
    args = ap.parse_args()

    K = 6
    M = 3 * K + 2
    env = DemixingEnv(K=K, Nf=3, Ninf=128, provide_hint=True,
                      provide_influence=False, N_stations=args.stations,
                      seed=args.seed)
    buf = TrainingBuffer(args.samples, n_input=M, n_output=K - 1)
    for i in range(args.samples):
        obs = env.reset()
        hint = env.get_hint()
        buf.store(obs["metadata"].reshape(-1), hint[:K - 1])
        print(f"{i + 1}/{args.samples} hint={np.round(hint[:K - 1], 2)}")
        buf.save_checkpoint(args.out)


if __name__ == "__main__":
    main()
