#!/bin/bash
# round-2 call C: probe variant sweep + PMC counters for the pass1 /
# probe kernels + retest the fixed plan test
{ timeout 500 python tools/probe_sweep2.py; echo "SWEEP_RC=$?"; } \
  > gpurun_out/r02c_sweep.log 2>&1
{ timeout 300 python -m pytest tests/test_gpu_plan.py tests/test_gpu_sparse.py -q; \
  echo "PYTEST_RC=$?"; } > gpurun_out/r02c_pytest.log 2>&1
export TMPDIR=/tmp
cd /tmp
R=/root/repo
rm -rf /tmp/pf /tmp/pw && mkdir -p /tmp/pf /tmp/pw
{ timeout 400 rocprofv3 --pmc FETCH_SIZE -d /tmp/pf -o f -- \
    python $R/bench.py --steps 2 --warmup 1 --skip-cpu-baseline > /dev/null;
  echo FETCH_RC=$?;
  timeout 400 rocprofv3 --pmc WRITE_SIZE -d /tmp/pw -o w -- \
    python $R/bench.py --steps 2 --warmup 1 --skip-cpu-baseline > /dev/null;
  echo WRITE_RC=$?; } > $R/gpurun_out/r02c_pmc.log 2>&1
FDB=$(find /tmp/pf -name '*.db' | head -1); WDB=$(find /tmp/pw -name '*.db' | head -1)
{ echo "== FETCH_SIZE"; python $R/tools/rocpd_summary.py "$FDB" FETCH_SIZE;
  echo "== WRITE_SIZE"; python $R/tools/rocpd_summary.py "$WDB" WRITE_SIZE; } \
  > $R/gpurun_out/r02c_pmc_summary.txt 2>&1
cd $R
grep -E "RC=|us/launch|passed|failed" gpurun_out/r02c_sweep.log \
  gpurun_out/r02c_pytest.log gpurun_out/r02c_pmc.log
