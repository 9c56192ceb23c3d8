"""Train the MLP regressor metadata → selection probabilities.

Reference `demixing_rl/train_regressor.py`: 3-layer MLP (M→32→32→K−1
tanh) on (metadata, hint) pairs.
"""

import argparse
import sys
from pathlib import Path

import numpy as np
import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))

from smartcal_amd.models import RegressorNet, TrainingBuffer
from smartcal_amd.utils.device import default_device


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--buffer", default="databuffer.npy")
    ap.add_argument("--iters", default=5000, type=int)
    ap.add_argument("--batch", default=32, type=int)
    ap.add_argument("--out", default="regressor.model")
    ap.add_argument("--seed", default=0, type=int)
    args = ap.parse_args()
    torch.manual_seed(args.seed)
    np.random.seed(args.seed)
    device = default_device()

    K = 6
    M = 3 * K + 2
    buf = TrainingBuffer(1, M, K - 1)
    buf.load_checkpoint(args.buffer)
    net = RegressorNet(M, K - 1).to(device)
    opt = torch.optim.Adam(net.parameters(), lr=1e-3)
    for ci in range(args.iters):
        x, y = buf.sample(args.batch)
        xt = torch.from_numpy(x).to(device)
        yt = torch.from_numpy(y).to(device)
        opt.zero_grad()
        loss = torch.nn.functional.mse_loss(net(xt), yt)
        loss.backward()
        opt.step()
        if ci % 500 == 0:
            print(f"{ci} {float(loss):.5f}")
    torch.save(net.state_dict(), args.out)


if __name__ == "__main__":
    main()
