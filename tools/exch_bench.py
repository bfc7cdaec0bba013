#!/usr/bin/env python3
"""Q3 Motion-exchange overhead at SF100: local path vs the full
redistribute path forced through real RCCL (world-1 self-loopback,
GG_FORCE_EXCHANGE) — estimates the per-step collective+sync cost the
8-GPU SCALE run pays."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from greengage_amd import Engine, PGDate
from greengage_amd.engine import PIPE_Q1, PIPE_Q3

eng = Engine(device=0, n_segments=1, segment_id=0)
eng.comm_init(eng.comm_id())
li = eng.register_synth("lineitem", seed=42, sf=100)
od = eng.register_synth("orders", seed=42, sf=100)
cu = eng.register_synth("customer", seed=42, sf=100)
p3 = eng.compile(PIPE_Q3, lineitem=li, orders=od, customer=cu,
                 cutoff_date=PGDate("1995-03-15"), mktsegment=2, limit_k=10)

base = eng.execute_q3(p3)
t0 = time.perf_counter()
for _ in range(8):
    r = eng.execute_q3(p3)
local = (time.perf_counter() - t0) / 8
assert r == base

os.environ["GG_FORCE_EXCHANGE"] = "1"
r2 = eng.execute_q3(p3)
assert r2 == base, "loopback parity"
t0 = time.perf_counter()
for _ in range(8):
    r2 = eng.execute_q3(p3)
exch = (time.perf_counter() - t0) / 8
assert r2 == base
st = {s["name"]: s for s in eng.stats(p3)}
print(f"q3 local {local*1e3:.2f} ms/step; forced-exchange {exch*1e3:.2f} "
      f"ms/step; exchange overhead {(exch-local)*1e3:.2f} ms")
print("orders_exchange stat:", round(st["orders_exchange"]["total_ms"]
      / st["orders_exchange"]["launches"], 3), "ms/launch")

# Q1 combine leg
p1 = eng.compile(PIPE_Q1, lineitem=li, cutoff_date=PGDate("1998-08-15"))
os.environ.pop("GG_FORCE_EXCHANGE")
g = eng.execute_q1(p1)
t0 = time.perf_counter()
for _ in range(8):
    eng.execute_q1(p1)
l1 = (time.perf_counter() - t0) / 8
os.environ["GG_FORCE_EXCHANGE"] = "1"
g2 = eng.execute_q1(p1)
assert g == g2
t0 = time.perf_counter()
for _ in range(8):
    eng.execute_q1(p1)
e1 = (time.perf_counter() - t0) / 8
print(f"q1 local {l1*1e3:.2f}; with combine allgather {e1*1e3:.2f}; "
      f"overhead {(e1-l1)*1e3:.2f} ms")
print("EXCH_BENCH_OK")
