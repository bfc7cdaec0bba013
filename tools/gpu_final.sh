#!/bin/bash
# comprehensive final validation: smoke, full GPU suite, SF100 full-size
# parity, default bench (incl. cpu_baseline leg)
cd /root/repo
{ timeout 300 python __graft_entry__.py smoke; echo "SMOKE_RC=$?";
  timeout 600 python -m pytest tests -m gpu -q; echo "PYTEST_RC=$?";
  GG_BIG=1 timeout 600 python -m pytest tests/test_gpu_fullsize.py -m gpu -q -x; echo "BIG_RC=$?";
  timeout 500 python bench.py --steps 5 --warmup 2; echo "BENCH_RC=$?";
} > gpurun_out/final.log 2>&1
grep -E "RC=|passed|failed|smoke ok" gpurun_out/final.log
