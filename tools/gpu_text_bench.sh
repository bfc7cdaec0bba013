#!/bin/bash
# text-path rates measured at the C-ABI boundary
cd /root/repo
timeout 600 python - <<'PY' > gpurun_out/text_bench.log 2>&1
import ctypes, sys, time
sys.path.insert(0, "."); sys.path.insert(0, "oracle"); sys.path.insert(0, "tests")
import numpy as np, pyoracle
from greengage_amd import Engine
from greengage_amd.engine import lib, I64

n = 20_000_000
rng = np.random.default_rng(9)
segs = [b"AUTOMOBILE", b"BUILDING", b"FURNITURE", b"HOUSEHOLD", b"MACHINERY"]
codes0 = rng.integers(0, 5, n)
vals = [segs[c] for c in codes0]
nulls = np.zeros(n, np.uint8)
framed, nb = pyoracle.dsb_encode_text(vals, nulls, 2, 1, blocksz=32768)
ao = np.ascontiguousarray(pyoracle.ao_wrap(framed), np.uint8)
print(f"ao bytes {len(ao):,} blocks {nb}")

eng = Engine(device=0, n_segments=1, segment_id=0)
L = lib()
offs = np.empty(n + 10, np.uint64); lens = np.empty(n + 10, np.uint32)
onulls = np.empty(n + 10, np.uint8); pool = np.empty(len(ao) * 8 + (1<<16), np.uint8)
cnt, plen = I64(), I64()
vp = ctypes.c_void_p
for trial in range(3):
    t0 = time.time()
    rc = L.gg_engine_aocs_decode_ao_text(
        ao.ctypes.data_as(vp), len(ao), 1, 2, 2, 0,
        offs.ctypes.data_as(vp), lens.ctypes.data_as(vp),
        onulls.ctypes.data_as(vp), n + 10,
        pool.ctypes.data_as(vp), len(pool), ctypes.byref(cnt), ctypes.byref(plen))
    t1 = time.time()
    assert rc == 0 and cnt.value == n
    print(f"trial {trial}: text decode ABI {t1-t0:.3f}s = "
          f"{len(ao)/1e9/(t1-t0):.2f} GB/s of AO bytes, {n/(t1-t0)/1e6:.0f} M rows/s")

codes = np.zeros(n, np.int32)
dbytes = np.zeros(1 << 20, np.uint8); doffs = np.zeros(4097, np.int64)
nd = ctypes.c_int32()
for trial in range(3):
    t2 = time.time()
    rc = L.gg_engine_text_dict_encode(
        pool.ctypes.data_as(vp), offs.ctypes.data_as(vp),
        lens.ctypes.data_as(vp), onulls.ctypes.data_as(vp), n, 4096,
        codes.ctypes.data_as(vp), dbytes.ctypes.data_as(vp), 1 << 20,
        doffs.ctypes.data_as(vp), ctypes.byref(nd))
    t3 = time.time()
    assert rc == 0 and nd.value == 5
    print(f"trial {trial}: dict encode ABI {t3-t2:.3f}s = "
          f"{n/(t3-t2)/1e6:.0f} M rows/s")
d = [bytes(dbytes[doffs[i]:doffs[i+1]]) for i in range(5)]
assert d == sorted(segs)
exp = np.array([d.index(segs[c]) for c in range(5)])[codes0]
assert np.array_equal(codes, exp)
print("parity OK")
eng.shutdown()
PY
tail -10 gpurun_out/text_bench.log
