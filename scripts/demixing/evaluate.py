"""Deployment-style evaluation: fresh observation → model recommendation.

Reference `demixing/evaluate.py:20-58` (real-MS glob + time window →
get_info_from_dataset → forward → recommendation); the data source here
is a fresh in-memory simulation.
"""

import argparse
import sys
from pathlib import Path

import numpy as np
import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))

from smartcal_amd.models import TransformerEncoder
from smartcal_amd.radio.dataset import generate_training_example
from smartcal_amd.utils.device import default_device


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="transformer.model")
    ap.add_argument("--ninf", default=64, type=int)
    ap.add_argument("--seed", default=0, type=int)
    ap.add_argument("--stations", default=62, type=int)
    args = ap.parse_args()
    device = default_device()
    rng = np.random.default_rng(args.seed)

    K = 6
    Nout = args.ninf * args.ninf + 8
    x, y, _ = generate_training_example(rng, Ninf=args.ninf,
                                        N_stations=args.stations)
    net = TransformerEncoder(num_layers=1, input_dim=K * Nout,
                             model_dim=K * (args.ninf + 2),
                             num_classes=K - 1, num_heads=K).to(device)
    sd = torch.load(args.model, map_location=device, weights_only=True)
    net.load_state_dict(sd["model_state_dict"])
    with torch.no_grad():
        probs = net(torch.from_numpy(x[None]).to(device))[0].cpu().numpy()
    names = ["CasA", "CygA", "HerA", "TauA", "VirA"]
    print("recommendation (probability of demixing each outlier):")
    for nm, p, t in zip(names, probs, y):
        print(f"  {nm}: {p:.3f} (truth {int(t)})")


if __name__ == "__main__":
    main()
