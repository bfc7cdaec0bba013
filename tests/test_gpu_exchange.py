"""Motion-exchange plumbing on one GPU: the same partition → RCCL
alltoallv → probe → redistribute → insert path that runs at N>1,
exercised as a world-size-1 self-loopback (GG_FORCE_EXCHANGE=1), must
produce bit-identical Q3 results to the fused single-GPU path."""
import os

import pytest

import pyoracle

pytestmark = pytest.mark.gpu


def test_q3_exchange_loopback_bitexact():
    from greengage_amd import Engine, PGDate
    from greengage_amd.engine import PIPE_Q3

    eng = Engine(device=0, n_segments=1, segment_id=0)
    try:
        eng.comm_init(eng.comm_id())    # RCCL world of 1
        li = eng.register_synth("lineitem", seed=42, sf=1)
        od = eng.register_synth("orders", seed=42, sf=1)
        cu = eng.register_synth("customer", seed=42, sf=1)
        cutoff = PGDate("1995-03-15")

        p_fused = eng.compile(PIPE_Q3, lineitem=li, orders=od, customer=cu,
                              cutoff_date=cutoff, mktsegment=2, limit_k=10)
        rows_f, hdr_f = eng.execute_q3(p_fused)

        os.environ["GG_FORCE_EXCHANGE"] = "1"
        try:
            p_x = eng.compile(PIPE_Q3, lineitem=li, orders=od, customer=cu,
                              cutoff_date=cutoff, mktsegment=2, limit_k=10)
            rows_x, hdr_x = eng.execute_q3(p_x)
            # steady state (scratch reuse, cached sizing): same again
            rows_x2, hdr_x2 = eng.execute_q3(p_x)
        finally:
            del os.environ["GG_FORCE_EXCHANGE"]

        assert hdr_x == hdr_f
        assert rows_x == rows_f
        assert hdr_x2 == hdr_f and rows_x2 == rows_f

        # and the whole thing against the CPU oracle
        topk, res = pyoracle.q3_synth(42, 1, cutoff)
        assert hdr_f["n_groups"] == res["n_groups"]
        assert hdr_f["group_checksum"] == res["group_checksum"]
        assert rows_f == topk
    finally:
        eng.shutdown()


def test_q1_allgather_loopback():
    """Q1's 2-stage-agg combine over RCCL allgather, world-1 loopback."""
    import os
    from greengage_amd import Engine, PGDate
    from greengage_amd.engine import PIPE_Q1

    eng = Engine(device=0, n_segments=1, segment_id=0)
    try:
        eng.comm_init(eng.comm_id())
        li = eng.register_synth("lineitem", seed=42, sf=1)
        p = eng.compile(PIPE_Q1, lineitem=li,
                        cutoff_date=PGDate("1998-08-15"))
        base = eng.execute_q1(p)
        os.environ["GG_FORCE_EXCHANGE"] = "1"
        try:
            assert eng.execute_q1(p) == base
        finally:
            del os.environ["GG_FORCE_EXCHANGE"]
    finally:
        eng.shutdown()


def test_q1_steady_state_reuse():
    """Repeated executes on one pipeline (scratch reuse) stay
    bit-identical — the bench measures exactly this steady state."""
    from greengage_amd import Engine, PGDate
    from greengage_amd.engine import PIPE_Q1, PIPE_Q3

    eng = Engine(device=0, n_segments=1, segment_id=0)
    try:
        li = eng.register_synth("lineitem", seed=42, sf=1)
        od = eng.register_synth("orders", seed=42, sf=1)
        cu = eng.register_synth("customer", seed=42, sf=1)
        p1 = eng.compile(PIPE_Q1, lineitem=li,
                         cutoff_date=PGDate("1998-08-15"))
        a = eng.execute_q1(p1)
        for _ in range(3):
            assert eng.execute_q1(p1) == a
        p3 = eng.compile(PIPE_Q3, lineitem=li, orders=od, customer=cu,
                         cutoff_date=PGDate("1995-03-15"), mktsegment=2)
        r0 = eng.execute_q3(p3)
        for _ in range(3):
            assert eng.execute_q3(p3) == r0
    finally:
        eng.shutdown()


def test_q5_exchange_loopback_bitexact():
    """Q5's broadcast-Motion (supplier) + two redistribute legs as a
    world-1 self-loopback must equal the fused path."""
    import os
    from greengage_amd import Engine, PGDate
    from greengage_amd.engine import PIPE_Q5

    eng = Engine(device=0, n_segments=1, segment_id=0)
    try:
        eng.comm_init(eng.comm_id())
        li = eng.register_synth("lineitem", seed=42, sf=1)
        od = eng.register_synth("orders", seed=42, sf=1)
        cu = eng.register_synth("customer", seed=42, sf=1)
        su = eng.register_synth("supplier", seed=42, sf=1)
        na = eng.register_synth("nation", seed=42, sf=1)
        lo, hi = PGDate("1997-01-01"), PGDate("1998-01-01")
        p = eng.compile(PIPE_Q5, lineitem=li, orders=od, customer=cu,
                        supplier=su, nation=na, cutoff_date=lo,
                        cutoff_hi=hi, regionkey=1)
        rows_f = eng.execute_q5(p)
        os.environ["GG_FORCE_EXCHANGE"] = "1"
        try:
            p_x = eng.compile(PIPE_Q5, lineitem=li, orders=od, customer=cu,
                              supplier=su, nation=na, cutoff_date=lo,
                              cutoff_hi=hi, regionkey=1)
            rows_x = eng.execute_q5(p_x)
            rows_x2 = eng.execute_q5(p_x)
        finally:
            del os.environ["GG_FORCE_EXCHANGE"]
        assert rows_x == rows_f
        assert rows_x2 == rows_f
        expect = __import__("pyoracle").q5_rows(
            __import__("pyoracle").q5_synth(42, 1, 1, lo, hi))
        assert [(r["nationkey"], r["revenue4"], r["count"])
                for r in rows_f] == \
            [(r["nationkey"], r["revenue4"], r["count"]) for r in expect]
    finally:
        eng.shutdown()
