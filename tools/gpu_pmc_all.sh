#!/bin/bash
# PMC traffic (FETCH_SIZE/WRITE_SIZE) for ALL pipeline kernels (Q1+Q3+Q5)
export TMPDIR=/tmp
cd /tmp
R=/root/repo
rm -rf /tmp/pf /tmp/pw && mkdir -p /tmp/pf /tmp/pw
{ timeout 400 rocprofv3 --pmc FETCH_SIZE -d /tmp/pf -o f -- python $R/bench.py --steps 2 --warmup 1 --skip-cpu-baseline > /dev/null;
  echo FETCH_RC=$?;
  timeout 400 rocprofv3 --pmc WRITE_SIZE -d /tmp/pw -o w -- python $R/bench.py --steps 2 --warmup 1 --skip-cpu-baseline > /dev/null;
  echo WRITE_RC=$?; } > $R/gpurun_out/pmc_all.log 2>&1
FDB=$(find /tmp/pf -name '*.db' | head -1); WDB=$(find /tmp/pw -name '*.db' | head -1)
{ echo "== FETCH_SIZE (KB units, x2 correction per microarch guide applies to wide coalesced reads)";
  python $R/tools/rocpd_summary.py "$FDB" FETCH_SIZE;
  echo "== WRITE_SIZE";
  python $R/tools/rocpd_summary.py "$WDB" WRITE_SIZE; } > $R/gpurun_out/pmc_all_summary.txt 2>&1
grep -E "RC=" $R/gpurun_out/pmc_all.log; tail -40 $R/gpurun_out/pmc_all_summary.txt
