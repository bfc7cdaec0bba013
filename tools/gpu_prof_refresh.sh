#!/bin/bash
# refresh trace + PMC evidence for the FINAL kernel set
export TMPDIR=/tmp
cd /tmp
R=/root/repo
rm -rf /tmp/tr2 /tmp/pf2 && mkdir -p /tmp/tr2 /tmp/pf2
{ timeout 400 rocprofv3 --kernel-trace --stats -d /tmp/tr2 -o t -- python $R/bench.py --steps 3 --warmup 1 --skip-cpu-baseline > /dev/null;
  echo TRACE_RC=$?;
  timeout 400 rocprofv3 --pmc FETCH_SIZE -d /tmp/pf2 -o f -- python $R/bench.py --steps 2 --warmup 1 --skip-cpu-baseline > /dev/null;
  echo FETCH_RC=$?; } > $R/gpurun_out/prof_refresh.log 2>&1
TDB=$(find /tmp/tr2 -name '*.db' | head -1); FDB=$(find /tmp/pf2 -name '*.db' | head -1)
python $R/tools/rocpd_summary.py "$TDB" > $R/gpurun_out/kernel_trace_refresh.txt 2>&1
python $R/tools/rocpd_summary.py "$FDB" FETCH_SIZE > $R/gpurun_out/pmc_fetch_refresh.txt 2>&1
grep -E "RC=" $R/gpurun_out/prof_refresh.log; head -20 $R/gpurun_out/kernel_trace_refresh.txt
