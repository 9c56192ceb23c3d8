"""Generate (x, y) training data for the demixing transformer classifier.

Reference `demixing/simulate_data.py:40-56`: loop
generate_training_data(Ninf=64) into a supervised buffer. Here each
example is generated fully in memory (`radio.dataset`).
"""

import argparse
import sys
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))

from smartcal_amd.models import SupervisedBuffer
from smartcal_amd.radio.dataset import generate_training_example


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--samples", default=100, type=int)
    ap.add_argument("--ninf", default=64, type=int)
    ap.add_argument("--stations", default=26, type=int)
    ap.add_argument("--seed", default=0, type=int)
    ap.add_argument("--out", default="simul_data.buffer")
    args = ap.parse_args()
    rng = np.random.default_rng(args.seed)
    K = 6
    Nout = args.ninf * args.ninf + 8
    buf = SupervisedBuffer(args.samples, (K * Nout,), (K - 1,))
    for i in range(args.samples):
        x, y, _ = generate_training_example(rng, Ninf=args.ninf,
                                            N_stations=args.stations)
        buf.store_data(x, y)
        print(f"{i + 1}/{args.samples} labels={y}")
    buf.save_checkpoint(args.out)


if __name__ == "__main__":
    main()
