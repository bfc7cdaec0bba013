#!/bin/bash
# fresh kernel-trace profile of the full bench (Q1+Q3+Q5) + SF300 evidence
cd /tmp && export TMPDIR=/tmp
rm -rf /tmp/proff && mkdir -p /tmp/proff
{ timeout 600 rocprofv3 --kernel-trace --stats -d /tmp/proff -o tr -- python /root/repo/bench.py --steps 3 --warmup 1 --skip-cpu-baseline > /tmp/bench_prof.json 2>/tmp/bench_prof.err; echo "PROF_RC=$?"; } > /root/repo/gpurun_out/proffinal.log 2>&1
DB=$(find /tmp/proff -name '*.db' | head -1)
{ python /root/repo/tools/rocpd_summary.py "$DB"; } > /root/repo/gpurun_out/kernel_trace_final.txt 2>&1
cp /tmp/bench_prof.json /root/repo/gpurun_out/ 2>/dev/null
cd /root/repo
{ timeout 600 python bench.py --sf 300 --steps 3 --warmup 1 --skip-cpu-baseline > gpurun_out/sf300_final.json 2>gpurun_out/sf300_final.err; echo "SF300_RC=$?"; } >> gpurun_out/proffinal.log 2>&1
{ timeout 300 python -m pytest tests/test_gpu_engine.py tests/test_gpu_fullsize.py -q; echo "PYTEST_RC=$?"; } >> gpurun_out/proffinal.log 2>&1
grep -E "RC=|passed|failed" gpurun_out/proffinal.log
