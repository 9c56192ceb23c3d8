"""Compare MLP regressor vs TSK fuzzy net vs the data-driven hint.

Reference `demixing_rl/evaluate_tsk_msp.py:62-86`: evaluate each
predictor's action on live env steps and report rewards.
"""

import argparse
import sys
from pathlib import Path

import numpy as np
import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))

from smartcal_amd.envs.demix import DemixingEnv
from smartcal_amd.models import RegressorNet, TSKModel


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--episodes", default=3, type=int)
    ap.add_argument("--stations", default=26, type=int)
    ap.add_argument("--regressor", default="regressor.model")
    ap.add_argument("--tsk", default="tsk.model")
    ap.add_argument("--rules", default=3, type=int)
    ap.add_argument("--seed", default=0, type=int)
    args = ap.parse_args()

    K = 6
    M = 3 * K + 2
    env = DemixingEnv(K=K, Nf=3, Ninf=128, provide_hint=True,
                      provide_influence=False, N_stations=args.stations,
                      seed=args.seed)
    reg = RegressorNet(M, K - 1)
    try:
        reg.load_state_dict(torch.load(args.regressor,
                                       weights_only=True))
    except FileNotFoundError:
        reg = None
    tsk = TSKModel(M, K - 1, n_rule=args.rules)
    try:
        tsk.load_state_dict(torch.load(args.tsk, weights_only=True))
    except FileNotFoundError:
        tsk = None

    META_SCALE = 1e3
    xmean = np.zeros(M, np.float32)
    xmean[0:5] = 64
    xmean[12:18] = 30
    xmean[18] = 20
    xmean[19] = 50
    xmean /= META_SCALE

    scores = {"mlp": [], "tsk": [], "hint": []}
    for _ in range(args.episodes):
        obs = env.reset()
        md = obs["metadata"].reshape(-1)
        hint = env.get_hint()
        preds = {"hint": hint}
        for name, model in (("mlp", reg), ("tsk", tsk)):
            if model is None:
                continue
            with torch.no_grad():
                p = model(torch.from_numpy((md - xmean)[None]))[0].numpy()
            a = np.zeros(K, np.float32)
            a[:K - 1] = p
            a[K - 1] = hint[K - 1]
            preds[name] = a
        for name, a in preds.items():
            _, r, *_ = env.step(np.asarray(a, np.float32))
            scores[name].append(float(r))
    for name, rs in scores.items():
        if rs:
            print(f"{name}: mean reward {np.mean(rs):.4f}")


if __name__ == "__main__":
    main()
