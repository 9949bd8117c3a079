#!/bin/bash
# Round-1 validation pass 2: full GPU suite, bench.py (overlapped cold start),
# refreshed rocprof stats for profiles/.
set -x
cd /root/repo
L=gpurun_out/validate2.log
echo "=== pytest -m gpu (full) ===" > $L
timeout 600 python -m pytest tests/ -q -m gpu 2>&1 | tail -4 >> $L
echo "rc=$?" >> $L

echo "=== bench.py 1 GPU ===" >> $L
timeout 900 python bench.py --gpus 1 --steps 10 --warmup 3 >> $L 2>&1
echo "bench rc=$?" >> $L

echo "=== rocprof kernel stats (refresh) ===" >> $L
cd /tmp && export TMPDIR=/tmp
timeout 400 rocprofv3 --kernel-trace --stats --output-format csv \
  -d /root/repo/gpurun_out/prof3 -o trace -- \
  python /root/repo/scripts/profile_workload.py >> $L 2>&1
echo "rocprof rc=$?" >> $L
cd /root/repo
tail -c 6000 $L
