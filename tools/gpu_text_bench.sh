#!/bin/bash
# text-path rates: AO text decode + dictionary encode at scale
cd /root/repo
timeout 600 python - <<'PY' > gpurun_out/text_bench.log 2>&1
import sys, time
sys.path.insert(0, "."); sys.path.insert(0, "oracle"); sys.path.insert(0, "tests")
import numpy as np, pyoracle
from greengage_amd import Engine
from greengage_amd.engine import Engine as E

n = 20_000_000
rng = np.random.default_rng(9)
segs = [b"AUTOMOBILE", b"BUILDING", b"FURNITURE", b"HOUSEHOLD", b"MACHINERY"]
codes0 = rng.integers(0, 5, n)
vals = [segs[c] for c in codes0]
nulls = np.zeros(n, np.uint8)
t0 = time.time()
framed, nb = pyoracle.dsb_encode_text(vals, nulls, 2, 1, blocksz=32768)
ao = pyoracle.ao_wrap(framed)
t1 = time.time()
print(f"ref text encode+wrap {t1-t0:.1f}s, ao bytes {len(ao):,} blocks {nb}")

eng = Engine(device=0, n_segments=1, segment_id=0)
for trial in range(3):
    t2 = time.time()
    gv, gn = E.aocs_decode_ao_text(ao, 1, 2, 2, n + 10)
    t3 = time.time()
    print(f"trial {trial}: text decode {t3-t2:.3f}s = "
          f"{len(ao)/1e9/(t3-t2):.2f} GB/s, {n/(t3-t2)/1e6:.0f} M rows/s")
assert gv[0] == vals[0] and gv[n-1] == vals[n-1]
for trial in range(3):
    t4 = time.time()
    codes, d = E.text_dict_encode(gv, gn)
    t5 = time.time()
    print(f"trial {trial}: dict encode {t5-t4:.3f}s = "
          f"{n/(t5-t4)/1e6:.0f} M rows/s, dict {len(d)}")
assert d == sorted(segs)
assert np.array_equal(np.array([d.index(segs[c]) for c in range(5)])[codes0], codes)
print("parity OK")
eng.shutdown()
PY
tail -10 gpurun_out/text_bench.log
