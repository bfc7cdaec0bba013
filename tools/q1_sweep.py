#!/usr/bin/env python3
"""Q1 kernel tuning sweep on one MI355X: grid size x nontemporal loads.
Reports per-launch kernel ms (engine HIP events) for SF100."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from greengage_amd import Engine, PGDate
from greengage_amd.engine import PIPE_Q1

eng = Engine(device=0, n_segments=1, segment_id=0)
li = eng.register_synth("lineitem", seed=42, sf=100)
cutoff = PGDate("1998-08-15")
base = None
for nt in (0, 1):
    for grid in (1024, 2048, 4096, 8192, 16384):
        os.environ["GG_Q1_GRID"] = str(grid)
        if nt:
            os.environ["GG_Q1_NT"] = "1"
        else:
            os.environ.pop("GG_Q1_NT", None)
        p = eng.compile(PIPE_Q1, lineitem=li, cutoff_date=cutoff)
        r = eng.execute_q1(p)          # warmup
        for _ in range(5):
            r = eng.execute_q1(p)
        if base is None:
            base = r
        assert r == base, (nt, grid)   # parity across variants
        st = {s["name"]: s for s in eng.stats(p)}["q1_agg"]
        ms = st["total_ms"] / st["launches"]
        gbs = 600_000_000 * 38 / (ms / 1000) / 1e9
        print(f"nt={nt} grid={grid:6d}: {ms:.3f} ms/launch  {gbs:7.0f} GB/s")
eng.shutdown()
print("parity held across all variants")
