#!/bin/bash
# round-2 certification: full suite + long bench + trace + PMC + smoke
{ timeout 1100 python -m pytest tests -m gpu -q; echo "PYTEST_RC=$?"; } \
  > gpurun_out/r02z_pytest.log 2>&1
{ timeout 500 python bench.py --steps 30 --warmup 5; echo "BENCH_RC=$?"; } \
  > gpurun_out/r02z_bench.log 2>&1
{ timeout 300 python tools/plan_bench.py; echo "PB_RC=$?"; } \
  > gpurun_out/r02z_planbench.log 2>&1
{ timeout 200 python -c "import __graft_entry__ as g; g.smoke()"; \
  echo "SM_RC=$?"; } > gpurun_out/r02z_smoke.log 2>&1
bash tools/gpu_spill_bench.sh
cp gpurun_out/spill_bench.log gpurun_out/r02z_spill.log
export TMPDIR=/tmp
cd /tmp
R=/root/repo
rm -rf /tmp/proff /tmp/pf && mkdir -p /tmp/proff /tmp/pf
{ timeout 400 rocprofv3 --kernel-trace --stats -d /tmp/proff -o tr -- \
    python $R/bench.py --steps 3 --warmup 1 --skip-cpu-baseline \
    > /dev/null 2>&1; echo "TRACE_RC=$?";
  timeout 400 rocprofv3 --pmc FETCH_SIZE -d /tmp/pf -o f -- \
    python $R/bench.py --steps 2 --warmup 1 --skip-cpu-baseline \
    > /dev/null 2>&1; echo "PMC_RC=$?"; } > $R/gpurun_out/r02z_prof.log 2>&1
DB=$(find /tmp/proff -name '*.db' | head -1)
FDB=$(find /tmp/pf -name '*.db' | head -1)
python $R/tools/rocpd_summary.py "$DB" > $R/gpurun_out/r02z_kernel_trace.txt 2>&1
python $R/tools/rocpd_summary.py "$FDB" FETCH_SIZE > $R/gpurun_out/r02z_pmc_fetch.txt 2>&1
cd $R
grep -E "passed|failed|RC=" gpurun_out/r02z_pytest.log gpurun_out/r02z_prof.log
grep -oE "\"value\": [0-9.]+|\"ms_per_step\": [0-9.]+|\"q3_ms_per_step\": [0-9.]+|\"q5_ms_per_step\": [0-9.]+|\"frac\": [0-9.]+" gpurun_out/r02z_bench.log | head -5
tail -5 gpurun_out/r02z_planbench.log; tail -1 gpurun_out/r02z_smoke.log
tail -6 gpurun_out/r02z_spill.log
