#!/bin/bash
cd /root/repo
{ echo "=== one-pass ==="
  timeout 300 python bench.py --steps 3 --warmup 1 --skip-cpu-baseline 2>/dev/null | python3 -c "
import json,sys
for l in sys.stdin:
    if l.startswith('{\"metric\"'):
        d=json.loads(l); x=d['extra']
        q5p=[k for k in x['q5_kernel_stats'] if 'probe' in k['name']][0]
        print('q5', round(x['q5_ms_per_step'],3), 'probe', round(q5p['total_ms']/q5p['launches'],3), 'out', x['q5_n_out'])
"
  echo "=== two-pass ==="
  GG_Q5_TWOPASS=1 timeout 300 python bench.py --steps 3 --warmup 1 --skip-cpu-baseline 2>/dev/null | python3 -c "
import json,sys
for l in sys.stdin:
    if l.startswith('{\"metric\"'):
        d=json.loads(l); x=d['extra']
        q5p=[k for k in x['q5_kernel_stats'] if 'probe' in k['name']][0]
        print('q5', round(x['q5_ms_per_step'],3), 'probe', round(q5p['total_ms']/q5p['launches'],3), 'out', x['q5_n_out'])
"
  GG_Q5_TWOPASS=1 timeout 400 python -m pytest tests/test_gpu_engine.py tests/test_gpu_fullsize.py -q 2>&1 | tail -2
} > gpurun_out/q5twopass.log 2>&1
cat gpurun_out/q5twopass.log
