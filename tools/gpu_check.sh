#!/bin/bash
# runs on the GPU box: full gpu pytest + bench (no cpu baseline)
{ timeout 500 python -m pytest tests -m gpu -q; echo "PYTEST_RC=$?";
  timeout 400 python bench.py --steps 5 --warmup 2 --skip-cpu-baseline;
  echo "BENCH_RC=$?"; } > gpurun_out/gpu_check.log 2>&1
grep -E "passed|failed|RC=" gpurun_out/gpu_check.log
