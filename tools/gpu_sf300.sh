#!/bin/bash
{ timeout 900 python bench.py --sf 300 --steps 3 --warmup 1;
  echo "SF300_RC=$?"; } > gpurun_out/sf300.log 2>&1
grep -E "SF300_RC" gpurun_out/sf300.log
