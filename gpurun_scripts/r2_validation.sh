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
# --stats prints the kernel summary table on exit; that stdout IS the
# small committable artifact (the sqlite DBs under /tmp/prof stay on-box)
rocprofv3 --kernel-trace --stats -d /tmp/prof -- \
  python /root/repo/bench.py --steps 100 --warmup 20 \
  > /root/repo/gpurun_out/r2_kernel_stats.txt 2>&1
tail -5 /root/repo/gpurun_out/r2_kernel_stats.txt
