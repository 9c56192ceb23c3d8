"""Kernel-time summary from a rocprofv3 rocpd SQLite results DB.

rocprofv3 on this image writes `<out>_results.db` (rocpd schema) under
the -d dir. Usage: python gpurun_scripts/kstats.py <dir-or-db> [> out.txt]
Prints kernel / calls / tot_ms / avg_us / pct sorted by total time.
"""

import glob
import os
import sqlite3
import sys


def main():
    p = sys.argv[1]
    db = p
    if os.path.isdir(p):
        cands = sorted(glob.glob(f"{p}/**/*results.db", recursive=True),
                       key=os.path.getmtime)
        if not cands:
            print(f"no *results.db under {p}", file=sys.stderr)
            sys.exit(1)
        db = cands[-1]
    con = sqlite3.connect(db)
    cur = con.cursor()
    tables = [r[0] for r in cur.execute(
        "SELECT name FROM sqlite_master WHERE type IN ('table','view')")]
    disp = [t for t in tables if t.startswith("rocpd_kernel_dispatch")]
    sym = [t for t in tables if t.startswith("rocpd_info_kernel_symbol")]
    if not disp or not sym:
        print(f"rocpd tables not found in {db}: {tables}", file=sys.stderr)
        sys.exit(1)
    rows = cur.execute(f"""
        SELECT ks.display_name, COUNT(*), SUM(k.end - k.start) / 1e6,
               AVG(k.end - k.start) / 1e3
        FROM {disp[0]} k JOIN {sym[0]} ks ON k.kernel_id = ks.id
        GROUP BY ks.display_name ORDER BY 3 DESC""").fetchall()
    total = sum(r[2] for r in rows) or 1.0
    ndisp = sum(r[1] for r in rows)
    print(f"{'kernel':<72} {'calls':>6} {'tot_ms':>8} {'avg_us':>7} "
          f"{'pct':>5}")
    nrows = int(os.environ.get("KSTATS_ROWS", "40"))
    for name, calls, ms, us in rows[:nrows]:
        print(f"{name[:72]:<72} {calls:>6} {ms:>8.2f} {us:>7.1f} "
              f"{100 * ms / total:>5.1f}")
    print(f"TOTAL kernel-ms: {total:.1f} over {ndisp} dispatches")


if __name__ == "__main__":
    main()
