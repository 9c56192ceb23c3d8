"""Plot the trained TSK model's membership functions and responses.

Parity with `demixing_rl/plot_tsk.py`.
"""

import argparse
import sys
from pathlib import Path

import numpy as np
import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))

from smartcal_amd.models import TSKModel


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--tsk", default="tsk.model")
    ap.add_argument("--rules", default=3, type=int)
    ap.add_argument("--out", default="tsk_memberships.png")
    args = ap.parse_args()
    K = 6
    M = 3 * K + 2
    model = TSKModel(M, K - 1, n_rule=args.rules)
    model.load_state_dict(torch.load(args.tsk, weights_only=True))
    centers = model.antecedent.center.detach().numpy()    # (M, R)
    sigmas = model.antecedent.sigma.detach().numpy()
    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt
    nshow = min(M, 8)
    fig, axes = plt.subplots(nshow, 1, figsize=(6, 2 * nshow))
    xs = np.linspace(-2, 2, 200)
    for d in range(nshow):
        axm = axes[d] if nshow > 1 else axes
        for r in range(args.rules):
            mf = np.exp(-(xs - centers[d, r]) ** 2
                        / (2 * sigmas[d, r] ** 2 + 1e-12))
            axm.plot(xs, mf, label=f"rule {r}")
        axm.set_ylabel(f"in {d}")
    plt.tight_layout()
    plt.savefig(args.out, dpi=120)
    print(f"wrote {args.out}")


if __name__ == "__main__":
    main()
