#!/usr/bin/env python3
"""Round-2 probe/top-k variant sweep on one MI355X at SF100.

Variants (GG_Q3_PROBE_VAR / GG_Q5_PROBE_VAR):
  0 strided 4-way unroll, 8 blocks/CU    2 quad 16-B loads, 8 blocks/CU
  1 quad 16-B loads, 4 blocks/CU         3 strided 4-way, 4 blocks/CU
Reports per-launch kernel ms from the engine's HIP-event stats; parity
asserted across variants.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from greengage_amd import Engine, PGDate
from greengage_amd.engine import PIPE_Q3, PIPE_Q5

eng = Engine(device=0, n_segments=1, segment_id=0)
li = eng.register_synth("lineitem", seed=42, sf=100)
od = eng.register_synth("orders", seed=42, sf=100)
cu = eng.register_synth("customer", seed=42, sf=100)
su = eng.register_synth("supplier", seed=42, sf=100)
na = eng.register_synth("nation", seed=42, sf=100)

base3 = base5 = None
for var in (0, 4, 5, 2):
    os.environ["GG_Q3_PROBE_VAR"] = str(var)
    os.environ["GG_Q5_PROBE_VAR"] = str(var)
    p3 = eng.compile(PIPE_Q3, lineitem=li, orders=od, customer=cu,
                     cutoff_date=PGDate("1995-03-15"), mktsegment=2,
                     limit_k=10)
    r3 = eng.execute_q3(p3)
    before = {s["name"]: dict(s) for s in eng.stats(p3)}
    for _ in range(6):
        r3 = eng.execute_q3(p3)
    if base3 is None:
        base3 = r3
    assert r3 == base3, f"Q3 variant {var} parity"
    after = {s["name"]: s for s in eng.stats(p3)}
    for k in ("probe_lineitem", "build_orders", "q3_topk"):
        d = ((after[k]["total_ms"] - before[k]["total_ms"])
             / (after[k]["launches"] - before[k]["launches"]))
        print(f"q3 var={var} {k}: {d*1000:.0f} us/launch", flush=True)

    p5 = eng.compile(PIPE_Q5, lineitem=li, orders=od, customer=cu,
                     supplier=su, nation=na,
                     cutoff_date=PGDate("1997-01-01"),
                     cutoff_hi=PGDate("1998-01-01"), regionkey=1)
    r5 = eng.execute_q5(p5)
    before = {s["name"]: dict(s) for s in eng.stats(p5)}
    for _ in range(6):
        r5 = eng.execute_q5(p5)
    if base5 is None:
        base5 = r5
    assert r5 == base5, f"Q5 variant {var} parity"
    after = {s["name"]: s for s in eng.stats(p5)}
    d = ((after["probe_lineitem_q5"]["total_ms"]
          - before["probe_lineitem_q5"]["total_ms"])
         / (after["probe_lineitem_q5"]["launches"]
            - before["probe_lineitem_q5"]["launches"]))
    print(f"q5 var={var} probe: {d*1000:.0f} us/launch", flush=True)
eng.shutdown()
print("SWEEP_OK")
