"""Plot the collected (metadata, hint) pairs.

Parity with `demixing_rl/plot_databuffer.py`.
"""

import argparse
import sys
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))

from smartcal_amd.models import TrainingBuffer


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--buffer", default="databuffer.npy")
    ap.add_argument("--out", default="databuffer.png")
    args = ap.parse_args()
    K = 6
    buf = TrainingBuffer(1, 3 * K + 2, K - 1)
    buf.load_checkpoint(args.buffer)
    n = min(buf.mem_cntr, buf.mem_size)
    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt
    fig, ax = plt.subplots(2, 1, figsize=(8, 6))
    ax[0].imshow(buf.x_[:n].T, aspect="auto", cmap="coolwarm")
    ax[0].set_title("metadata (inputs)")
    ax[1].imshow(buf.y_[:n].T, aspect="auto", cmap="coolwarm")
    ax[1].set_title("hints (targets)")
    plt.tight_layout()
    plt.savefig(args.out, dpi=120)
    print(f"wrote {args.out} ({n} samples)")


if __name__ == "__main__":
    main()
