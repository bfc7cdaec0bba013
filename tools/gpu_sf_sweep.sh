#!/bin/bash
cd /root/repo
{ for sf in 200 500; do
    echo "=== SF$sf ==="
    timeout 500 python bench.py --sf $sf --steps 2 --warmup 1 --skip-cpu-baseline 2>&1 | grep -E '^\{"metric"' 
    echo "RC=$?"
  done
} > gpurun_out/sf_sweep.log 2>&1
python3 - <<'PY'
import json
for l in open('gpurun_out/sf_sweep.log'):
    l=l.strip()
    if l.startswith('==') or l.startswith('RC'):
        print(l)
    elif l.startswith('{"metric"'):
        d=json.loads(l); x=d.get('extra',{})
        print('  Q1', round(d['ms_per_step'],2),'ms', round(d['value']/1e9,1),'Grows/s | Q3', round(x.get('q3_ms_per_step',0),2), '| Q5', round(x.get('q5_ms_per_step',0),2))
PY
