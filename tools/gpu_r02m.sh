#!/bin/bash
# round-2 call A: full GPU suite (incl. new multiproc/sparse tests) +
# bench with the v2 probe/topk kernels + kernel trace
{ timeout 1100 python -m pytest tests -m gpu -q; echo "PYTEST_RC=$?"; } \
  > gpurun_out/r02m_pytest.log 2>&1
{ timeout 400 python bench.py --steps 10 --warmup 3; echo "BENCH_RC=$?"; } \
  > gpurun_out/r02m_bench.log 2>&1
cd /tmp && export TMPDIR=/tmp
rm -rf /tmp/proff && mkdir -p /tmp/proff
{ timeout 400 rocprofv3 --kernel-trace --stats -d /tmp/proff -o tr -- \
    python /root/repo/bench.py --steps 3 --warmup 1 --skip-cpu-baseline \
    > /tmp/bench_prof.json 2>/tmp/bench_prof.err; echo "PROF_RC=$?"; } \
  > /root/repo/gpurun_out/r02m_prof.log 2>&1
DB=$(find /tmp/proff -name '*.db' | head -1)
python /root/repo/tools/rocpd_summary.py "$DB" \
  > /root/repo/gpurun_out/r02m_kernel_trace.txt 2>&1
cd /root/repo
grep -E "passed|failed|RC=" gpurun_out/r02m_pytest.log gpurun_out/r02m_prof.log
tail -c 1500 gpurun_out/r02m_bench.log
