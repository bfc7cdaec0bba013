#!/bin/bash
export TMPDIR=/tmp
R=$GRAFT_REPO_ROOT
cd $R
{ timeout 500 rocprofv3 --kernel-trace -d $R/gpurun_out/q3prof -- python bench.py --steps 3 --warmup 1 --skip-cpu-baseline; echo RC=$?; } > gpurun_out/q3prof.log 2>&1
grep RC= gpurun_out/q3prof.log
