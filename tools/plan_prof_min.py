#!/usr/bin/env python3
"""Minimal grouped-plan run for rocprofv3: one Q1-shaped descriptor,
bake engages after the first execute, then 4 more executes of the
baked-fast kernel."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from greengage_amd import Engine, PGDate
from greengage_amd.engine import NEG_INF

eng = Engine(device=0, n_segments=1, segment_id=0)
li = eng.register_synth("lineitem", seed=42, sf=100)
cutoff = PGDate("1998-08-15")
p = eng.compile_plan(
    li, preds=[("shipdate", NEG_INF, cutoff + 1)],
    group_cols=["rflag", "lstatus"],
    aggs=["count", ("sum", [("qty", "id")]), ("sum", [("price", "id")]),
          ("sum", [("disc", "id")]),
          ("sum", [("price", "id"), ("disc", "sub100")]),
          ("sum", [("price", "id"), ("disc", "sub100"),
                   ("tax", "add100")])])
g0 = eng.execute_plan(p, max_groups=16)
for _ in range(4):
    assert eng.execute_plan(p, max_groups=16) == g0
print("PLAN_PROF_OK",
      [s["name"] for s in eng.stats(p) if s["name"].startswith("path")])
eng.shutdown()
