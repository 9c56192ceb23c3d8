"""Tile the image states stored in a CNN replay buffer to a PNG.

Parity with `calibration/inspect_replaybuffer.py:20-26`.
"""

import argparse
import sys
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))

from smartcal_amd.rl.buffers_dict import DictReplayBuffer


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--buffer", default="replaymem_cnn.model")
    ap.add_argument("--out", default="replay_states.png")
    ap.add_argument("--ncols", default=8, type=int)
    ap.add_argument("--max", default=64, type=int)
    args = ap.parse_args()

    buf = DictReplayBuffer(1, (1, 128, 128), 20, 6)
    buf.load_checkpoint(args.buffer)
    n = min(len(buf), args.max)
    imgs = buf.img_memory[:n, 0].cpu().numpy()
    H, W = imgs.shape[1:]
    rows = (n + args.ncols - 1) // args.ncols
    tile = np.zeros((rows * H, args.ncols * W), np.float32)
    for i in range(n):
        r, c = divmod(i, args.ncols)
        tile[r * H:(r + 1) * H, c * W:(c + 1) * W] = imgs[i]
    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt
    plt.figure(figsize=(args.ncols, rows))
    plt.imshow(tile, cmap="viridis")
    plt.axis("off")
    plt.savefig(args.out, dpi=120, bbox_inches="tight")
    print(f"wrote {args.out} ({n} states)")


if __name__ == "__main__":
    main()
