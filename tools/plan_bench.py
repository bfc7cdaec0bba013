#!/usr/bin/env python3
"""Generic-descriptor perf check at SF100: Q6 via compile_plan.
The generic scan-agg kernel should stream its 4 predicate columns +
2 agg factors near roofline (28 B/row algorithmic)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from greengage_amd import Engine, pgdate

NEG_INF = -(1 << 63)

eng = Engine(device=0, n_segments=1, segment_id=0)
li = eng.register_synth("lineitem", seed=42, sf=100)
lo, hi = pgdate(1994, 1, 1), pgdate(1995, 1, 1)
p = eng.compile_plan(
    li,
    preds=[("shipdate", lo, hi), ("disc", 5, 8), ("qty", NEG_INF, 2400)],
    aggs=[("sum", [("price", "id"), ("disc", "id")]), "count"])
g0 = eng.execute_plan(p, max_groups=8)
before = {s["name"]: dict(s) for s in eng.stats(p)}
t0 = time.perf_counter()
steps = 10
for _ in range(steps):
    g = eng.execute_plan(p, max_groups=8)
wall = (time.perf_counter() - t0) / steps
assert g == g0
after = {s["name"]: s for s in eng.stats(p)}
d = ((after["plan_scan_agg"]["total_ms"] - before["plan_scan_agg"]["total_ms"])
     / (after["plan_scan_agg"]["launches"] - before["plan_scan_agg"]["launches"]))
rows = 600_000_000
print(f"q6 plan_scan_agg: {d*1000:.0f} us/launch = "
      f"{rows/ d / 1e6:.0f} M rows/ms... {rows/(d/1e3)/1e9:.1f} G rows/s, "
      f"{rows*28/ (d/1e3) / 1e12:.2f} TB/s algorithmic (28 B/row)")
print(f"q6 wall {wall*1e3:.2f} ms/exec; count={g0[0][2][1]} rev={g0[0][2][0]}")
print("PLAN_BENCH_OK")

# grouped generic plan (Q1-shaped: 6 char1-pair groups, 6 aggs) at SF100
from greengage_amd import PGDate
cutoff = PGDate("1998-08-15")
pq1 = eng.compile_plan(
    li, preds=[("shipdate", NEG_INF, cutoff + 1)],
    group_cols=["rflag", "lstatus"],
    aggs=["count", ("sum", [("qty", "id")]), ("sum", [("price", "id")]),
          ("sum", [("disc", "id")]),
          ("sum", [("price", "id"), ("disc", "sub100")]),
          ("sum", [("price", "id"), ("disc", "sub100"), ("tax", "add100")])])
gq0 = eng.execute_plan(pq1, max_groups=16)
before = {s["name"]: dict(s) for s in eng.stats(pq1)}
t0 = time.perf_counter()
for _ in range(5):
    gq = eng.execute_plan(pq1, max_groups=16)
wall = (time.perf_counter() - t0) / 5
assert gq == gq0
after = {s["name"]: s for s in eng.stats(pq1)}
d = ((after["plan_scan_agg"]["total_ms"] - before["plan_scan_agg"]["total_ms"])
     / (after["plan_scan_agg"]["launches"] - before["plan_scan_agg"]["launches"]))
print(f"q1-shaped grouped plan_scan_agg: {d*1000:.0f} us/launch "
      f"({600e6*38/(d/1e3)/1e12:.2f} TB/s algorithmic 38 B/row); "
      f"wall {wall*1e3:.2f} ms/exec")
print("PLAN_BENCH2_OK")

# NT-load variant of the generic kernel (GG_PLAN_NT=1)
os.environ["GG_PLAN_NT"] = "1"
gnt = eng.execute_plan(p, max_groups=8)
assert gnt == g0
before = {s["name"]: dict(s) for s in eng.stats(p)}
t0 = time.perf_counter()
for _ in range(10):
    gnt = eng.execute_plan(p, max_groups=8)
after = {s["name"]: s for s in eng.stats(p)}
d = ((after["plan_scan_agg"]["total_ms"] - before["plan_scan_agg"]["total_ms"])
     / (after["plan_scan_agg"]["launches"] - before["plan_scan_agg"]["launches"]))
print(f"q6 NT plan_scan_agg: {d*1000:.0f} us/launch")
gq = eng.execute_plan(pq1, max_groups=16)
assert gq == gq0
before = {s["name"]: dict(s) for s in eng.stats(pq1)}
for _ in range(5):
    gq = eng.execute_plan(pq1, max_groups=16)
after = {s["name"]: s for s in eng.stats(pq1)}
d = ((after["plan_scan_agg"]["total_ms"] - before["plan_scan_agg"]["total_ms"])
     / (after["plan_scan_agg"]["launches"] - before["plan_scan_agg"]["launches"]))
print(f"q1-shaped NT grouped: {d*1000:.0f} us/launch")
os.environ.pop("GG_PLAN_NT")
print("PLAN_BENCH3_OK")

# interpreted-vs-RTC comparison (fresh pipelines; RTC decision is per
# pipeline at first execute)
os.environ["GG_PLAN_RTC"] = "0"
p_i = eng.compile_plan(
    li, preds=[("shipdate", lo, hi), ("disc", 5, 8), ("qty", NEG_INF, 2400)],
    aggs=[("sum", [("price", "id"), ("disc", "id")]), "count"])
gi = eng.execute_plan(p_i, max_groups=8)
assert gi == g0, "interp/rtc parity"
pq_i = eng.compile_plan(
    li, preds=[("shipdate", NEG_INF, cutoff + 1)],
    group_cols=["rflag", "lstatus"],
    aggs=["count", ("sum", [("qty", "id")]), ("sum", [("price", "id")]),
          ("sum", [("disc", "id")]),
          ("sum", [("price", "id"), ("disc", "sub100")]),
          ("sum", [("price", "id"), ("disc", "sub100"), ("tax", "add100")])])
gqi = eng.execute_plan(pq_i, max_groups=16)
assert gqi == gq0, "interp/rtc grouped parity"
os.environ.pop("GG_PLAN_RTC")
for name, pp in (("q6", p_i), ("q1-shaped", pq_i)):
    before = {s["name"]: dict(s) for s in eng.stats(pp)}
    for _ in range(5):
        eng.execute_plan(pp, max_groups=16)
    after = {s["name"]: s for s in eng.stats(pp)}
    d = ((after["plan_scan_agg"]["total_ms"]
          - before["plan_scan_agg"]["total_ms"])
         / (after["plan_scan_agg"]["launches"]
            - before["plan_scan_agg"]["launches"]))
    print(f"{name} interp plan_scan_agg: {d*1000:.0f} us/launch")
for pp, name in ((p, "q6"), (pq1, "q1-shaped")):
    paths = [s["name"] for s in eng.stats(pp) if s["name"].startswith("path")]
    print(f"{name} default paths: {paths}")
print("PLAN_BENCH4_OK")

# LDS replica sweep for the RTC grouped kernel
for repl in (4, 8, 16, 32):
    os.environ["GG_PLAN_LREPL"] = str(repl)
    pr = eng.compile_plan(
        li, preds=[("shipdate", NEG_INF, cutoff + 1)],
        group_cols=["rflag", "lstatus"],
        aggs=["count", ("sum", [("qty", "id")]), ("sum", [("price", "id")]),
              ("sum", [("disc", "id")]),
              ("sum", [("price", "id"), ("disc", "sub100")]),
              ("sum", [("price", "id"), ("disc", "sub100"),
                       ("tax", "add100")])])
    gr = eng.execute_plan(pr, max_groups=16)
    assert gr == gq0
    before = {s["name"]: dict(s) for s in eng.stats(pr)}
    for _ in range(5):
        eng.execute_plan(pr, max_groups=16)
    after = {s["name"]: s for s in eng.stats(pr)}
    d = ((after["plan_scan_agg"]["total_ms"]
          - before["plan_scan_agg"]["total_ms"])
         / (after["plan_scan_agg"]["launches"]
            - before["plan_scan_agg"]["launches"]))
    print(f"grouped RTC LREPL={repl}: {d*1000:.0f} us/launch")
os.environ.pop("GG_PLAN_LREPL")
print("PLAN_BENCH5_OK")

# waves/SIMD sweep for the baked grouped kernel (GG_PLAN_WAVES)
for wv in (2, 3, 4):
    os.environ["GG_PLAN_WAVES"] = str(wv)
    pw = eng.compile_plan(
        li, preds=[("shipdate", NEG_INF, cutoff + 1)],
        group_cols=["rflag", "lstatus"],
        aggs=["count", ("sum", [("qty", "id")]), ("sum", [("price", "id")]),
              ("sum", [("disc", "id")]),
              ("sum", [("price", "id"), ("disc", "sub100")]),
              ("sum", [("price", "id"), ("disc", "sub100"),
                       ("tax", "add100")])])
    gw = eng.execute_plan(pw, max_groups=16)   # first: generic + bake
    assert gw == gq0
    before = {s["name"]: dict(s) for s in eng.stats(pw)}
    for _ in range(5):
        gw = eng.execute_plan(pw, max_groups=16)
    after = {s["name"]: s for s in eng.stats(pw)}
    d = ((after["plan_scan_agg"]["total_ms"]
          - before["plan_scan_agg"]["total_ms"])
         / (after["plan_scan_agg"]["launches"]
            - before["plan_scan_agg"]["launches"]))
    print(f"grouped baked WAVES={wv}: {d*1000:.0f} us/launch")
os.environ.pop("GG_PLAN_WAVES", None)
print("PLAN_BENCH6_OK")
