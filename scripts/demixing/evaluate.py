"""Deployment-style evaluation: observation → model recommendation.

Reference `demixing/evaluate.py:20-58` (real-MS glob + time window →
get_info_from_dataset → forward → recommendation). Data sources:

* ``--npz 'L_SB*.npz'`` — REAL observations exported to the npz bridge
  (`radio.ms_io.ms_to_npz` on a casacore machine, one file per
  sub-band), or a direct ``--ms`` glob when python-casacore is
  installed;
* default — a fresh in-memory simulation (no flags).
"""

import argparse
import glob as globmod
import sys
from pathlib import Path

import numpy as np
import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))

from smartcal_amd.models import TransformerEncoder
from smartcal_amd.radio.dataset import (generate_training_example,
                                        info_from_observation)
from smartcal_amd.utils.device import default_device


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="transformer.model")
    ap.add_argument("--ninf", default=64, type=int)
    ap.add_argument("--seed", default=0, type=int)
    ap.add_argument("--stations", default=62, type=int)
    ap.add_argument("--npz", default=None,
                    help="glob of exported sub-band npz files (real data)")
    ap.add_argument("--ms", default=None,
                    help="glob of MeasurementSets (needs python-casacore)")
    args = ap.parse_args()
    device = default_device()
    rng = np.random.default_rng(args.seed)

    K = 6
    Nout = args.ninf * args.ninf + 8
    if args.npz or args.ms:
        from smartcal_amd.radio import ms_io
        if args.npz:
            vis = ms_io.observation_from_npz(
                sorted(globmod.glob(args.npz)))
        else:
            vis = ms_io.merge_visdata(
                [ms_io.read_ms(p) for p in sorted(globmod.glob(args.ms))])
        x, _ = info_from_observation(vis, Ninf=args.ninf)
        y = np.full(K - 1, np.nan)   # truth unknown for real data
    else:
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
        truth = "?" if not np.isfinite(t) else str(int(t))
        print(f"  {nm}: {p:.3f} (truth {truth})")


if __name__ == "__main__":
    main()
