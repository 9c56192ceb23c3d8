#!/bin/bash
# Round-2 opening GPU validation (run via:
#   gpurun --timeout 900 -- 'bash gpurun_scripts/r2_validation.sh')
# 1. full GPU test suite (incl. the new batched/hint/KKT/transformer tests)
# 2. flagship bench
# 3. fresh kernel-stats profile after the guard/filter changes
set -x
python -m pytest tests/ -x -q -m gpu > gpurun_out/r2_tests.log 2>&1
echo "TESTS_RC=$?" >> gpurun_out/r2_tests.log
python bench.py --steps 300 --warmup 50 > gpurun_out/r2_bench.json 2>&1
echo "BENCH_RC=$?" >> gpurun_out/r2_bench.json
cd /tmp && export TMPDIR=/tmp
rocprofv3 --kernel-trace --stats -d /tmp/prof -- \
  python /root/repo/bench.py --steps 100 --warmup 20 \
  > /root/repo/gpurun_out/r2_prof_bench.log 2>&1
# summarize the rocpd sqlite DB to a small text file on-box
python - <<'PY' > /root/repo/gpurun_out/r2_kernel_stats.txt 2>&1
import glob, sqlite3, collections
dbs = glob.glob("/tmp/prof/**/*.db", recursive=True)
if not dbs:
    print("no rocprof db found")
else:
    con = sqlite3.connect(dbs[0])
    rows = con.execute(
        "select name, count(*), sum(end-start)/1e6 from rocpd_kernel_dispatch "
        "kd join rocpd_info_kernel_symbol ks on kd.kernel_id=ks.id "
        "group by name order by 3 desc limit 25").fetchall()
    tot = sum(r[2] for r in rows)
    for name, calls, ms in rows:
        print(f"{name[:68]:68s} {calls:6d} {ms:8.2f} {100*ms/tot:5.1f}%")
PY
