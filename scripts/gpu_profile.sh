#!/bin/bash
# rocprofv3 evidence collection: (1) kernel-trace + stats, (2) PMC counters
# (separate runs — --pmc must not be combined with trace domains on this pool).
set -x
cd /tmp && export TMPDIR=/tmp
OUT=/root/repo/gpurun_out/prof_r02
mkdir -p $OUT

timeout 400 rocprofv3 --kernel-trace --stats --output-format csv -d $OUT -o trace -- \
  python /root/repo/scripts/profile_workload.py > $OUT/trace.log 2>&1
echo "trace rc=$?" >> $OUT/trace.log

timeout 400 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_VALU_MFMA_BUSY_CYCLES \
  SQ_LDS_BANK_CONFLICT SQ_LDS_IDX_ACTIVE SQ_WAIT_ANY SQ_WAIT_INST_ANY \
  SQ_ACTIVE_INST_ANY GRBM_GUI_ACTIVE \
  --output-format csv -d $OUT -o pmc -- \
  python /root/repo/scripts/profile_workload.py > $OUT/pmc.log 2>&1
echo "pmc rc=$?" >> $OUT/pmc.log

ls -la $OUT
# summarize the stats csv (top kernels)
for f in $OUT/*stats*.csv; do echo "== $f =="; head -30 "$f"; done
for f in $OUT/*counter_collection*.csv; do
  python /root/repo/scripts/pmc_summarize.py "$f" $OUT/pmc_summary_r02.json &&     cat $OUT/pmc_summary_r02.json
done
tail -5 $OUT/trace.log $OUT/pmc.log
