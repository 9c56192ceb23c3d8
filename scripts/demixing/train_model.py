"""Train the demixing transformer classifier.

Reference `demixing/train_model.py:39-84`: TransformerEncoder
(num_layers=1, input_dim=K·(Ninf²+8), model_dim=K·(Ninf+2)... the
reference uses 6·66 for Ninf=64) + BCELoss + Adam closure loop, saving
{'model_state_dict': ...} checkpoints.
"""

import argparse
import sys
from pathlib import Path

import numpy as np
import torch
import torch.nn as nn

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))

from smartcal_amd.models import SupervisedBuffer, TransformerEncoder
from smartcal_amd.utils.device import default_device


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--buffer", default="simul_data.buffer")
    ap.add_argument("--iters", default=32000, type=int)
    ap.add_argument("--batch", default=64, type=int)
    ap.add_argument("--ninf", default=64, type=int)
    ap.add_argument("--out", default="transformer.model")
    ap.add_argument("--seed", default=0, type=int)
    args = ap.parse_args()
    torch.manual_seed(args.seed)
    np.random.seed(args.seed)
    device = default_device()

    K = 6
    Nout = args.ninf * args.ninf + 8
    buf = SupervisedBuffer(1, (K * Nout,), (K - 1,))
    buf.load_checkpoint(args.buffer)
    net = TransformerEncoder(num_layers=1, input_dim=K * Nout,
                             model_dim=K * (args.ninf + 2),
                             num_classes=K - 1, num_heads=K,
                             dropout=0.6).to(device)
    criterion = nn.BCELoss()
    opt = torch.optim.Adam(net.parameters(), lr=1e-3)
    for ci in range(args.iters):
        x, y = buf.sample_minibatch(min(args.batch, buf.mem_cntr))
        xt = torch.from_numpy(x).to(device)
        yt = torch.from_numpy(y).to(device)

        def closure():
            if torch.is_grad_enabled():
                opt.zero_grad()
            loss = criterion(net(xt), yt)
            if loss.requires_grad:
                loss.backward()
            return loss

        loss = opt.step(closure)
        if ci % 100 == 0:
            print(f"{ci} {float(loss):.5f}")
            torch.save({"model_state_dict": net.state_dict()}, args.out)
    torch.save({"model_state_dict": net.state_dict()}, args.out)


if __name__ == "__main__":
    main()
