#!/bin/bash
cd /root/repo
EX='
import json,sys
for l in sys.stdin:
    if l.startswith("{\"metric\""):
        d=json.loads(l); x=d["extra"]
        ks=[k for k in x["q5_kernel_stats"] if "probe" in k["name"] or "gather" in k["name"]]
        print("q5", round(x["q5_ms_per_step"],3), {k["name"]: round(k["total_ms"]/k["launches"],3) for k in ks}, "out", x["q5_n_out"])
'
{ echo "=== one-pass (GG_Q5_ONEPASS=1) ==="
  GG_Q5_ONEPASS=1 timeout 300 python bench.py --steps 3 --warmup 1 --skip-cpu-baseline 2>/dev/null | python3 -c "$EX"
  echo "=== two-pass (default) ==="
  timeout 300 python bench.py --steps 3 --warmup 1 --skip-cpu-baseline 2>/dev/null | python3 -c "$EX"
  timeout 400 python -m pytest tests/test_gpu_engine.py tests/test_gpu_fullsize.py tests/test_gpu_exchange.py -q 2>&1 | tail -2
} > gpurun_out/q5twopass.log 2>&1
cat gpurun_out/q5twopass.log
