#!/bin/bash
# final certification: long Q1 timing + smoke + full suites + big parity
cd /root/repo
{ timeout 300 python bench.py --steps 30 --warmup 5 --skip-q3 --skip-cpu-baseline | tail -1; echo "LONG_RC=$?";
  timeout 300 python __graft_entry__.py smoke; echo "SMOKE_RC=$?";
  timeout 600 python -m pytest tests -m gpu -q; echo "PYTEST_RC=$?";
  GG_BIG=1 timeout 600 python -m pytest tests/test_gpu_fullsize.py -m gpu -q -x; echo "BIG_RC=$?";
  timeout 500 python bench.py --steps 5 --warmup 2; echo "BENCH_RC=$?";
} > gpurun_out/cert.log 2>&1
grep -E "RC=|passed|failed|smoke ok" gpurun_out/cert.log
python3 -c "
import json
ls=[l for l in open('gpurun_out/cert.log') if l.startswith('{\"metric\"')]
for l in ls:
    d=json.loads(l)
    print('Q1', d['steps'],'steps:', round(d['ms_per_step'],3),'ms', round(d['value']/1e9,1),'Grows/s frac', round(d['roofline']['frac'],3))"
