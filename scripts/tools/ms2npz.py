"""Export MeasurementSets to the npz interchange schema.

Run on a casacore-equipped machine (e.g. the LOFAR reduction node):

    python scripts/tools/ms2npz.py L123_SB001.MS [L123_SB002.MS ...]

writes one `<ms-stem>.npz` per input, readable here via
`radio.ms_io.observation_from_npz` (see docs/DEMO.md §4).
"""

import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))

from smartcal_amd.radio import ms_io


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("ms", nargs="+", help="MeasurementSet paths")
    ap.add_argument("--col", default="DATA",
                    help="data column (DATA, CORRECTED_DATA, MODEL_DATA)")
    ap.add_argument("--out-dir", default=".",
                    help="directory for the npz files")
    args = ap.parse_args()
    for p in args.ms:
        out = Path(args.out_dir) / (Path(p).stem + ".npz")
        ms_io.ms_to_npz(p, str(out), col=args.col)
        print(f"{p} -> {out}")


if __name__ == "__main__":
    main()
