#!/bin/bash
cd /root/repo
GG_SPILL_DEBUG=1 timeout 800 python - <<'PY' > gpurun_out/spill_bench.log 2>&1
import sys, time
sys.path.insert(0, ".")
import numpy as np
from greengage_amd import Engine
from greengage_amd.engine import Engine as E

eng = Engine(device=0, n_segments=1, segment_id=0)
n = 1_000_000_000
rng = np.random.default_rng(11)
keys = rng.integers(0, 5_000_000, n).astype(np.int64)
vals = rng.integers(-100, 100, n).astype(np.int64)
print(f"input {n:,} rows = {n*16/1e9:.0f} GB of pairs")
for budget, label in ((1 << 30, "1 GiB budget (spill, cold pool)"),
                      (1 << 30, "1 GiB budget (spill, warm pool)"),
                      (1 << 39, "in-memory (cold pool)"),
                      (1 << 39, "in-memory (warm pool)")):
    t0 = time.time()
    k, s, c, nparts = E.hash_groupby_spill(keys, vals, budget)
    t1 = time.time()
    print(f"{label}: {t1-t0:.2f}s = {n/(t1-t0)/1e6:.0f} M rows/s, "
          f"{nparts} partitions, {len(k):,} groups")
# cross-check the two paths
k1, s1, c1, p1 = E.hash_groupby_spill(keys[:50_000_000], vals[:50_000_000], 1 << 30)
k2, s2, c2, p2 = E.hash_groupby_spill(keys[:50_000_000], vals[:50_000_000], 1 << 39)
assert np.array_equal(k1, k2) and np.array_equal(s1, s2) and np.array_equal(c1, c2)
print("spill == in-memory on 50M subset: OK")
eng.shutdown()
PY
tail -16 gpurun_out/spill_bench.log
