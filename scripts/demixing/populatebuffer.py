"""Class-rebalance a supervised buffer by oversampling rare labels.

Parity with `demixing/populatebuffer.py:44-49` (the reference scaffolds
SMOTETomek from imblearn, which is not in this image; plain bootstrap
oversampling of minority label patterns is used instead).
"""

import argparse
import sys
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))

from smartcal_amd.models import SupervisedBuffer


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("buffer")
    ap.add_argument("--out", default="balanced.buffer")
    ap.add_argument("--seed", default=0, type=int)
    args = ap.parse_args()
    rng = np.random.default_rng(args.seed)
    b = SupervisedBuffer(1, (1,), (1,))
    b.load_checkpoint(args.buffer)
    n = min(b.mem_cntr, b.mem_size)
    X, Y = b.x[:n], b.y[:n]
    # group by label pattern, oversample all groups to the max count
    pats = [tuple(v) for v in (Y > 0.5).astype(int)]
    uniq = sorted(set(pats))
    groups = {u: [i for i, p in enumerate(pats) if p == u] for u in uniq}
    mx = max(len(g) for g in groups.values())
    out = SupervisedBuffer(mx * len(uniq), X.shape[1:], Y.shape[1:])
    for u, idx in groups.items():
        pick = rng.choice(idx, mx, replace=len(idx) < mx)
        for i in pick:
            out.store_data(X[i], Y[i])
    out.save_checkpoint(args.out)
    print(f"{len(uniq)} label patterns, {mx} each -> {args.out}")


if __name__ == "__main__":
    main()
