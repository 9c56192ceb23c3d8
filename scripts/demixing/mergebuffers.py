"""Merge two supervised data buffers (parity with
`demixing/mergebuffers.py`)."""

import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))

from smartcal_amd.models import SupervisedBuffer


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("a")
    ap.add_argument("b")
    ap.add_argument("--out", default="merged.buffer")
    args = ap.parse_args()
    ba = SupervisedBuffer(1, (1,), (1,))
    ba.load_checkpoint(args.a)
    bb = SupervisedBuffer(1, (1,), (1,))
    bb.load_checkpoint(args.b)
    na = min(ba.mem_cntr, ba.mem_size)
    nb = min(bb.mem_cntr, bb.mem_size)
    out = SupervisedBuffer(na + nb, ba.x.shape[1:], ba.y.shape[1:])
    for i in range(na):
        out.store_data(ba.x[i], ba.y[i])
    for i in range(nb):
        out.store_data(bb.x[i], bb.y[i])
    out.save_checkpoint(args.out)
    print(f"merged {na}+{nb} -> {args.out}")


if __name__ == "__main__":
    main()
