#!/bin/bash
# AO storage-path decode rate: 100M int64 rows through the full layer
cd /root/repo
timeout 600 python - <<'PY' > gpurun_out/ao_bench.log 2>&1
import sys, time
sys.path.insert(0, "."); sys.path.insert(0, "oracle"); sys.path.insert(0, "tests")
import numpy as np, pyoracle
from greengage_amd import Engine
from greengage_amd.engine import Engine as E

n = 100_000_000
rng = np.random.default_rng(5)
vals = rng.integers(-2**40, 2**40, n).astype(np.int64)
t0 = time.time()
framed, nb = pyoracle.dsb_encode(vals, np.zeros(n, np.uint8), 8, 2, 0, 0, blocksz=32768)
t1 = time.time()
ao = pyoracle.ao_wrap(framed)
t2 = time.time()
print(f"ref encode {t1-t0:.1f}s, wrap {t2-t1:.1f}s, ao bytes {len(ao):,} blocks {nb}")

eng = Engine(device=0, n_segments=1, segment_id=0)
for trial in range(3):
    t3 = time.time()
    gv, gn = E.aocs_decode_ao(ao, 1, 2, 2, 8, n + 10)
    t4 = time.time()
    gb = len(ao) / 1e9
    print(f"trial {trial}: decode_ao end-to-end {t4-t3:.3f}s = "
          f"{gb/(t4-t3):.2f} GB/s of AO bytes, {n/(t4-t3)/1e6:.0f} M rows/s")
assert np.array_equal(gv, vals)
print("bit-exact OK")

# device-direct mount (no host round trip for the results)
for trial in range(2):
    t5 = time.time()
    h = eng.register_table_ao(f"bench_ao_{trial}", [
        ("v", "int64", ao, 1, 2, 2, 0)])
    t6 = time.time()
    print(f"mount trial {trial}: register_table_ao {t6-t5:.3f}s = "
          f"{len(ao)/1e9/(t6-t5):.2f} GB/s")
eng.shutdown()
PY
cat gpurun_out/ao_bench.log | tail -10
