"""Evaluate the demixing transformer + influence-map sanity check.

Reference `demixing/eval_model.py:52-112`: briefly refit with LBFGS to
harvest curvature pairs, then `influence_matrix(net, x, y, optimizer)`
→ per-class per-direction influence values.
"""

import argparse
import sys
from pathlib import Path

import numpy as np
import torch
import torch.nn as nn

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))

from smartcal_amd.autograd_tools import influence_matrix
from smartcal_amd.models import SupervisedBuffer, TransformerEncoder
from smartcal_amd.optim.lbfgs import LBFGSNew
from smartcal_amd.utils.device import default_device


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--buffer", default="simul_data.buffer")
    ap.add_argument("--model", default="transformer.model")
    ap.add_argument("--ninf", default=64, type=int)
    ap.add_argument("--samples", default=16, type=int)
    args = ap.parse_args()
    device = default_device()

    K = 6
    Nout = args.ninf * args.ninf + 8
    buf = SupervisedBuffer(1, (K * Nout,), (K - 1,))
    buf.load_checkpoint(args.buffer)
    net = TransformerEncoder(num_layers=1, input_dim=K * Nout,
                             model_dim=K * (args.ninf + 2),
                             num_classes=K - 1, num_heads=K).to(device)
    sd = torch.load(args.model, map_location=device, weights_only=True)
    net.load_state_dict(sd["model_state_dict"])
    criterion = nn.BCELoss()

    # accuracy spot check
    n = min(args.samples, buf.mem_cntr)
    x = torch.from_numpy(buf.x[:n]).to(device)
    y = torch.from_numpy(buf.y[:n]).to(device)
    with torch.no_grad():
        pred = (net(x) > 0.5).float()
    acc = float((pred == y).float().mean())
    print(f"accuracy over {n} samples: {acc:.3f}")

    # harvest LBFGS curvature pairs then influence matrix
    opt = LBFGSNew(net.parameters(), history_size=7, max_iter=4,
                   line_search_fn=True, batch_mode=True)
    for _ in range(4):
        def closure():
            if torch.is_grad_enabled():
                opt.zero_grad()
            loss = criterion(net(x), y)
            if loss.requires_grad:
                loss.backward()
            return loss
        opt.step(closure)
    xi = x[:1].clone().requires_grad_(True)
    infl = influence_matrix(net, xi, y[:1], opt)
    print("influence matrix:", infl.shape,
          float(infl.abs().mean()))


if __name__ == "__main__":
    main()
