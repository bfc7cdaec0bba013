#!/bin/bash
export TMPDIR=/tmp
R=$GRAFT_REPO_ROOT
cd $R
{ timeout 600 rocprofv3 --kernel-trace --stats -d $R/gpurun_out/prof_full -- \
    python bench.py --steps 3 --warmup 1 --skip-cpu-baseline;
  echo "TRACE_RC=$?"; } > gpurun_out/prof_full.log 2>&1
{ timeout 300 rocprofv3 --kernel-trace -d $R/gpurun_out/prof_sort -- \
    python -m pytest tests/test_gpu_sort.py -m gpu -q;
  echo "SORT_RC=$?"; } >> gpurun_out/prof_full.log 2>&1
grep -E "RC=" gpurun_out/prof_full.log
