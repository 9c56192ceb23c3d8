"""Influence matrix of the trained TSK model.

Reference `demixing_rl/influence_tsk.py:64-72`: refit briefly with
LBFGS to harvest curvature pairs, then
`autograd_tools.influence_matrix`.
"""

import argparse
import sys
from pathlib import Path

import numpy as np
import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))

from smartcal_amd.autograd_tools import influence_matrix
from smartcal_amd.models import TrainingBuffer, TSKModel
from smartcal_amd.optim.lbfgs import LBFGSNew


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--buffer", default="databuffer.npy")
    ap.add_argument("--tsk", default="tsk.model")
    ap.add_argument("--rules", default=3, type=int)
    args = ap.parse_args()

    K = 6
    M = 3 * K + 2
    buf = TrainingBuffer(1, M, K - 1)
    buf.load_checkpoint(args.buffer)
    n = min(buf.mem_cntr, buf.mem_size)
    X = torch.from_numpy(buf.x_[:n])
    Y = torch.from_numpy(buf.y_[:n])
    model = TSKModel(M, K - 1, n_rule=args.rules)
    model.load_state_dict(torch.load(args.tsk, weights_only=True))

    opt = LBFGSNew(model.parameters(), history_size=7, max_iter=4,
                   line_search_fn=True, batch_mode=True)
    for _ in range(4):
        def closure():
            if torch.is_grad_enabled():
                opt.zero_grad()
            loss = ((model(X) - Y) ** 2).mean()
            if loss.requires_grad:
                loss.backward()
            return loss
        opt.step(closure)
    xi = X[:1].clone().requires_grad_(True)
    infl = influence_matrix(model, xi, Y[:1], opt)
    print("TSK influence matrix:", infl.shape)
    np.save("tsk_influence.npy", infl.detach().numpy())


if __name__ == "__main__":
    main()
