"""Format rocprofv3 --stats kernel CSV into the committed table form.

Usage (on the GPU box): python gpurun_scripts/kstats.py <dir> [> out.txt]
Finds the newest *kernel_stats.csv under <dir> and prints
kernel / calls / tot_ms / avg_us / pct sorted by total time.
"""

import csv
import glob
import sys


def main():
    d = sys.argv[1]
    files = sorted(glob.glob(f"{d}/**/*kernel_stats.csv", recursive=True))
    if not files:
        print(f"no kernel_stats.csv under {d}", file=sys.stderr)
        sys.exit(1)
    rows = []
    with open(files[-1]) as f:
        for r in csv.DictReader(f):
            name = r.get("Name") or r.get("NAME") or ""
            calls = int(r.get("Calls") or r.get("CALLS") or 0)
            dur = float(r.get("TotalDurationNs") or r.get("DURATION_NS")
                        or r.get("TOTAL_DURATION_NS") or 0)
            rows.append((name, calls, dur))
    total = sum(d for _, _, d in rows) or 1.0
    rows.sort(key=lambda x: -x[2])
    print(f"{'kernel':<70} {'calls':>6} {'tot_ms':>8} {'avg_us':>7} "
          f"{'pct':>5}")
    for name, calls, dur in rows[:40]:
        print(f"{name[:70]:<70} {calls:>6} {dur / 1e6:>8.2f} "
              f"{dur / 1e3 / max(calls, 1):>7.1f} {100 * dur / total:>5.1f}")


if __name__ == "__main__":
    main()
