#!/bin/bash
export TMPDIR=/tmp
R=$GRAFT_REPO_ROOT
cd $R
{ timeout 500 rocprofv3 --kernel-trace --stats -d $R/gpurun_out/p2_trace -- python bench.py --steps 3 --warmup 1 --skip-cpu-baseline;
  echo TRACE_RC=$?;
  timeout 300 rocprofv3 --pmc FETCH_SIZE -d $R/gpurun_out/p2_fetch -- python bench.py --steps 3 --warmup 1 --skip-q3 --skip-cpu-baseline;
  echo FETCH_RC=$?;
  timeout 300 rocprofv3 --pmc WRITE_SIZE -d $R/gpurun_out/p2_write -- python bench.py --steps 3 --warmup 1 --skip-q3 --skip-cpu-baseline;
  echo WRITE_RC=$?; } > gpurun_out/prof2.log 2>&1
grep -E "RC=" gpurun_out/prof2.log
