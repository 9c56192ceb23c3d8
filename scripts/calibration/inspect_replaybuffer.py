"""Tile the influence-map states of a pickled replay buffer into a PNG.

Reference `calibration/inspect_replaybuffer.py:1-26`: load the agent's
saved buffer and write a grid image of the stored state maps for visual
inspection. Works on the dict-buffer checkpoints written by
`rl/buffers_dict.py` (filled-prefix pickles).
"""

import argparse
import pickle
import sys
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("buffer", help="pickled buffer checkpoint "
                                   "(e.g. tmp/replaymem_sac.model)")
    ap.add_argument("--out", default="replaybuffer.png")
    ap.add_argument("--cols", type=int, default=8)
    ap.add_argument("--max", type=int, default=64,
                    help="max states to tile")
    args = ap.parse_args()

    with open(args.buffer, "rb") as f:
        sd = pickle.load(f)
    imgs = sd.get("img_memory")
    if imgs is None:
        raise SystemExit("no img_memory in checkpoint (MLP buffer?)")
    imgs = np.asarray(imgs)[: args.max]
    if imgs.ndim == 4:  # (n, C, H, W) → first channel
        imgs = imgs[:, 0]
    n, H, W = imgs.shape
    cols = min(args.cols, max(n, 1))
    rows = (n + cols - 1) // cols
    tile = np.zeros((rows * H, cols * W), dtype=np.float32)
    for i in range(n):
        r, c = divmod(i, cols)
        tile[r * H:(r + 1) * H, c * W:(c + 1) * W] = imgs[i]

    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt
    plt.figure(figsize=(cols, rows))
    plt.imshow(tile, cmap="viridis")
    plt.axis("off")
    plt.tight_layout()
    plt.savefig(args.out, dpi=120)
    print(f"wrote {args.out}: {n} states ({rows}x{cols} grid)")


if __name__ == "__main__":
    main()
