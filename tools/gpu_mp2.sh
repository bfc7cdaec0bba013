#!/bin/bash
# 2 engine ranks on ONE GPU.  RCCL refuses same-device communicators
# (profiles/r02_rccl_samedev.md), so the ranks exchange over the
# engine's shm transport (GG_COMM_SHM) — every engine-side multi-rank
# code path (partition/scatter, alltoallv layout, allgather combine)
# still executes for real.
export GG_COMM_SHM=1 GG_COMM_SHM_MB=512
{ timeout 900 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
    --master-addr 127.0.0.1 --master-port 29517 \
    bench.py --gpus 2 --steps 3 --warmup 1 --sf 10 --skip-cpu-baseline;
  echo "MP2_RC=$?"; } > gpurun_out/mp2.log 2>&1
grep -E "MP2_RC|Error|error|rows/s" gpurun_out/mp2.log | tail -8
