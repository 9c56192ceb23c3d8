"""Train the TSK fuzzy network on (metadata, hint) pairs.

Reference `demixing_rl/train_tsk.py:56-150`: 3 rules, order-1,
LayerNorm+ReLU-wrapped GMF antecedents, inverse center-distance + sigma
regularizers (g1=g2=1e-4), Adam closure loop, train/test split.
"""

import argparse
import sys
from pathlib import Path

import numpy as np
import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))

from smartcal_amd.models import (TrainingBuffer, TSKModel,
                                 center_difference_loss, sigma_loss)
from smartcal_amd.models.tsk import antecedent_init_center
from smartcal_amd.utils.device import default_device


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--buffer", default="databuffer.npy")
    ap.add_argument("--iters", default=20000, type=int)
    ap.add_argument("--batch", default=256, type=int)
    ap.add_argument("--rules", default=3, type=int)
    ap.add_argument("--out", default="tsk.model")
    ap.add_argument("--seed", default=0, type=int)
    args = ap.parse_args()
    torch.manual_seed(args.seed)
    np.random.seed(args.seed)
    device = default_device()

    K = 6
    M = 3 * K + 2
    buf = TrainingBuffer(1, M, K - 1)
    buf.load_checkpoint(args.buffer)
    X = buf.x_[:min(buf.mem_cntr, buf.mem_size)]
    Y = buf.y_[:min(buf.mem_cntr, buf.mem_size)]
    # reference mean-shift of the metadata (`train_tsk.py:37-44`)
    META_SCALE = 1e3
    xmean = np.zeros(M, np.float32)
    xmean[0:5] = 64
    xmean[12:18] = 30
    xmean[18] = 20
    xmean[19] = 50
    xmean /= META_SCALE
    X = X - xmean

    ntest = max(1, X.shape[0] // 10)
    x_test, y_test = X[:ntest], Y[:ntest]
    x_train, y_train = X[ntest:], Y[ntest:]

    model = TSKModel(M, K - 1, n_rule=args.rules,
                     init_center=antecedent_init_center(x_train,
                                                        args.rules)
                     ).to(device)
    opt = torch.optim.Adam(model.parameters(), lr=0.01)
    g1 = g2 = 1e-4
    bs = min(args.batch, x_train.shape[0])
    for ci in range(args.iters):
        idx = np.random.choice(x_train.shape[0], bs,
                               replace=x_train.shape[0] < bs)
        xt = torch.from_numpy(x_train[idx]).to(device)
        yt = torch.from_numpy(y_train[idx]).to(device)
        opt.zero_grad()
        loss = (model(xt) - yt).norm() ** 2 / bs \
            + g1 * center_difference_loss(model) + g2 * sigma_loss(model)
        loss.backward()
        opt.step()
        if ci % 500 == 0:
            with torch.no_grad():
                tl = float((model(torch.from_numpy(x_test).to(device))
                            - torch.from_numpy(y_test).to(device)
                            ).norm() ** 2 / ntest)
            print(f"{ci} train {float(loss):.5f} test {tl:.5f}")
    torch.save(model.state_dict(), args.out)


if __name__ == "__main__":
    main()
