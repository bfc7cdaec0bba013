#!/bin/bash
# full validation + fresh profiles + SF300 evidence
cd /root/repo
mkdir -p gpurun_out
{ timeout 600 python -m pytest tests -m gpu -q; echo "PYTEST_RC=$?"; } > gpurun_out/full2.log 2>&1
{ timeout 300 python bench.py --steps 5 --warmup 2 --skip-cpu-baseline; echo "BENCH_RC=$?"; } >> gpurun_out/full2.log 2>&1
cd /tmp && export TMPDIR=/tmp
{ timeout 600 rocprofv3 --kernel-trace --stats -d /tmp/prof -o trace -- python /root/repo/bench.py --steps 3 --warmup 1 --skip-cpu-baseline > /dev/null 2>&1; echo "PROF_RC=$?"; } >> /root/repo/gpurun_out/full2.log 2>&1
cp /tmp/prof/*trace*stats* /root/repo/gpurun_out/ 2>/dev/null || find /tmp/prof -name '*stats*' -exec cp {} /root/repo/gpurun_out/ \; 2>/dev/null
ls /tmp/prof >> /root/repo/gpurun_out/full2.log 2>&1
cd /root/repo
{ GG_BIG=300 timeout 500 python bench.py --steps 3 --warmup 1 --skip-cpu-baseline > gpurun_out/sf300.json 2>gpurun_out/sf300.err; echo "SF300_RC=$?"; } >> gpurun_out/full2.log 2>&1
grep -E "passed|failed|RC=" gpurun_out/full2.log
