#!/bin/bash
# size ceiling: Q1-only at SF600/SF700 (~3.6B / 4.2B rows)
cd /root/repo
{ for sf in 600 700; do
    echo "=== SF$sf (Q1 only) ==="
    timeout 420 python bench.py --sf $sf --steps 2 --warmup 1 --skip-q3 --skip-cpu-baseline 2>&1 | grep -E '^\{"metric"|error|Error' | head -3
    echo "RC=$?"
  done
} > gpurun_out/sf_max.log 2>&1
python3 - <<'PY'
import json
for l in open('gpurun_out/sf_max.log'):
    l=l.strip()
    if l.startswith('==') or l.startswith('RC'): print(l)
    elif l.startswith('{"metric"'):
        d=json.loads(l)
        print('  Q1', round(d['ms_per_step'],2),'ms', round(d['value']/1e9,1),'Grows/s')
    else: print(' ', l[:160])
PY
