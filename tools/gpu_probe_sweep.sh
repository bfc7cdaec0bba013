#!/bin/bash
# sweep probe grid sizes for Q3/Q5
cd /root/repo
for g in 2048 4096 8192 16384; do
  echo "=== GRID $g ==="
  GG_Q3_PROBE_GRID=$g GG_Q5_PROBE_GRID=$g timeout 300 python bench.py --steps 3 --warmup 1 --skip-cpu-baseline 2>/dev/null \
    | python3 -c "
import json,sys
for l in sys.stdin:
    if l.startswith('{\"metric\"'):
        d=json.loads(l); x=d['extra']
        q3p=[k for k in x['q3_kernel_stats'] if k['name']=='probe_lineitem'][0]
        q5p=[k for k in x['q5_kernel_stats'] if k['name']=='probe_lineitem_q5'][0]
        print('q3_probe', round(q3p['total_ms']/q3p['launches'],3), 'q5_probe', round(q5p['total_ms']/q5p['launches'],3), 'q3', round(x['q3_ms_per_step'],2), 'q5', round(x['q5_ms_per_step'],2))
"
done > gpurun_out/probe_sweep.log 2>&1
cat gpurun_out/probe_sweep.log
