#!/bin/bash
# 2 ranks on ONE GPU (both device 0): exercises the real multi-rank
# RCCL exchange paths if RCCL permits same-device communicators.
{ timeout 500 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
    --master-addr 127.0.0.1 --master-port 29517 \
    bench.py --gpus 2 --steps 3 --warmup 1 --sf 1 --skip-cpu-baseline;
  echo "MP2_RC=$?"; } > gpurun_out/mp2.log 2>&1
grep -E "MP2_RC|Error|error|rows/s" gpurun_out/mp2.log | tail -8
