"""Tie-heavy top-k: every group has the SAME revenue, so the sampled
threshold admits every group and the collect pass overflows its
first-try buffer — exercising the exact-size retry in exec_q3 (the
bounded-heap tie semantics of tuplesort.c:1360–1377: order among full
ties is refined by orderkey ASC on both sides)."""
import numpy as np
import pytest

import pyoracle

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def eng():
    from greengage_amd import Engine
    e = Engine(device=0, n_segments=1, segment_id=0)
    yield e
    e.shutdown()


def test_q3_all_groups_tied_retry_path(eng):
    from greengage_amd.engine import PIPE_Q3
    rng = np.random.default_rng(31)
    ncust, nord = 20_000, 100_000
    c_ck = np.arange(1, ncust + 1, dtype=np.int64)
    c_ms = np.full(ncust, 2, np.uint8)          # every customer matches
    o_ok = np.arange(1, nord + 1, dtype=np.int64)
    o_ck = rng.integers(1, ncust + 1, nord).astype(np.int64)
    o_od = np.full(nord, -1000, np.int32)       # all pass date < cutoff
    o_pr = np.zeros(nord, np.int32)
    nli = nord * 4
    l_ok = np.repeat(o_ok, 4)
    l_sd = np.full(nli, 500, np.int32)          # all pass shipdate > cutoff
    l_pc = np.full(nli, 100_000, np.int64)      # identical price
    l_dc = np.full(nli, 5, np.int64)            # identical discount

    cu = eng.register_table("tie_cust", [("custkey", "int64", c_ck),
                                         ("mktseg", "char1", c_ms)], ncust)
    od = eng.register_table("tie_ord", [("orderkey", "int64", o_ok),
                                        ("custkey", "int64", o_ck),
                                        ("orderdate", "int32", o_od),
                                        ("shippriority", "int32", o_pr)],
                            nord)
    li = eng.register_table("tie_li", [("orderkey", "int64", l_ok),
                                       ("shipdate", "int32", l_sd),
                                       ("price", "dec64", l_pc),
                                       ("disc", "dec64", l_dc)], nli)

    for k in (10, 1000):
        p = eng.compile(PIPE_Q3, lineitem=li, orders=od, customer=cu,
                        cutoff_date=0, mktsegment=2, limit_k=k)
        rows, hdr = eng.execute_q3(p, k=k)
        # 100k groups, all revenue = 4 * 100000 * 95 (scale 4)
        assert hdr["n_groups"] == nord
        assert hdr["n_join_rows"] == nli
        rev = 4 * 100_000 * 95
        assert hdr["rev_sum4"] == rev * nord
        assert len(rows) == k
        # full tie: winners are the k smallest orderkeys (date equal too)
        assert [r["orderkey"] for r in rows] == list(range(1, k + 1))
        assert all(r["revenue4"] == rev for r in rows)

    topk, res = pyoracle.q3_arrays(c_ck, c_ms, 2, o_ok, o_ck, o_od, o_pr,
                                   l_ok, l_sd, l_pc, l_dc, 0, k=10)
    p = eng.compile(PIPE_Q3, lineitem=li, orders=od, customer=cu,
                    cutoff_date=0, mktsegment=2, limit_k=10)
    rows, hdr = eng.execute_q3(p)
    assert hdr["group_checksum"] == res["group_checksum"]
    assert [(r["orderkey"], r["revenue4"]) for r in rows] == \
           [(r["orderkey"], r["revenue4"]) for r in topk]
