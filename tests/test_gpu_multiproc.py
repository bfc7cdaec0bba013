"""World>1 engine execution on ONE GPU.

RCCL refuses same-device communicator ranks (profiles/
r02_rccl_samedev.md), so these tests run N real engine processes on
one MI355X over the engine's shm transport (comm.cpp GG_COMM_SHM).
Everything BUT the byte transport is the production multi-rank path:
cdbhash sharding at generation, orders_filter/partition/scatter
kernels, alltoallv receive layout, dense-map insert of received rows,
allgather combine of partial states — asserted bit-exact against the
full-table CPU oracle.
"""
import json
import os
import subprocess
import sys

import pytest

import pyoracle

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
WORKER = os.path.join(REPO, "tests", "mp_engine_worker.py")

Q1_KEYS = ("count", "sum_qty_c", "sum_base_c", "sum_dcol_c", "sum_disc4",
           "sum_charge6")


def run_world(world, sf, tmp_path):
    env = dict(os.environ)
    env["GG_COMM_SHM"] = "1"
    env.setdefault("GG_COMM_SHM_MB", "192")
    idfile = str(tmp_path / f"commid.{world}")
    procs, outs = [], []
    for r in range(world):
        out = str(tmp_path / f"out.{world}.{r}.json")
        outs.append(out)
        procs.append(subprocess.Popen(
            [sys.executable, WORKER, str(r), str(world), idfile, out,
             str(sf)], env=env, cwd=REPO,
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
    fail = []
    for r, p in enumerate(procs):
        try:
            out, _ = p.communicate(timeout=600)
        except subprocess.TimeoutExpired:
            for q in procs:
                q.kill()
            raise
        if p.returncode != 0:
            fail.append(f"rank {r} rc={p.returncode}:\n"
                        f"{out.decode(errors='replace')[-2000:]}")
    assert not fail, "\n".join(fail)
    return [json.load(open(o)) for o in outs]


def check_results(results, world, sf):
    from greengage_amd import PGDate

    # shards partition the tables (no row lost or duplicated)
    assert sum(r["li_rows"] for r in results) == 6_000_000 * sf
    assert sum(r["od_rows"] for r in results) == 1_500_000 * sf

    # every rank holds the globally-combined result (allgather combine)
    for r in results[1:]:
        assert r["q1"] == results[0]["q1"]
        assert r["q3_hdr"] == results[0]["q3_hdr"]
        assert r["q3_rows"] == results[0]["q3_rows"]
        assert r["q5"] == results[0]["q5"]
        assert r["plan_q6"] == results[0]["plan_q6"]
        assert r["plan_grouped"] == results[0]["plan_grouped"]
        assert r["plan_grouped_repeat"] == results[0]["plan_grouped"]

    exp_q1 = [g for g in pyoracle.q1_synth(42, sf, PGDate("1998-08-15"))
              if g["count"]]
    got_q1 = results[0]["q1"]
    assert len(got_q1) == len(exp_q1)
    for got, exp in zip(got_q1, exp_q1):
        for k in Q1_KEYS:
            assert got[k] == exp[k], k

    exp_rows, exp_hdr = pyoracle.q3_synth(42, sf, PGDate("1995-03-15"), k=10)
    hdr = results[0]["q3_hdr"]
    assert hdr["n_groups"] == exp_hdr["n_groups"]
    assert hdr["n_join_rows"] == exp_hdr["n_join_rows"]
    assert hdr["rev_sum4"] == exp_hdr["rev_sum4"]
    assert hdr["group_checksum"] == exp_hdr["group_checksum"]
    assert [(r["orderkey"], r["revenue4"], r["orderdate"]) for r in
            results[0]["q3_rows"]] == \
           [(r["orderkey"], r["revenue4"], r["orderdate"]) for r in exp_rows]

    exp_q5 = pyoracle.q5_rows(pyoracle.q5_synth(
        42, sf, 1, PGDate("1997-01-01"), PGDate("1998-01-01")))
    assert [(r["nationkey"], r["count"], r["revenue4"]) for r in
            results[0]["q5"]] == \
           [(r["nationkey"], r["count"], r["revenue4"]) for r in exp_q5]

    # generic-descriptor plans combined across segments
    import numpy as np
    from greengage_amd import pgdate
    g = pyoracle.gen_lineitem(42, 0, 6_000_000 * sf)
    lo, hi = pgdate(1994, 1, 1), pgdate(1995, 1, 1)
    m = ((g["shipdate"] >= lo) & (g["shipdate"] < hi) & (g["disc"] >= 5)
         & (g["disc"] < 8) & (g["qty"] < 2400))
    exp_rev = sum(int(p) * int(d) for p, d in
                  zip(g["price"][m].tolist(), g["disc"][m].tolist()))
    q6 = results[0]["plan_q6"]
    assert q6[0][2][0] == exp_rev
    assert q6[0][2][1] == int(np.count_nonzero(m))

    cutoff = PGDate("1998-08-15")
    mg = g["shipdate"] <= cutoff
    grouped = results[0]["plan_grouped"]
    exp_g = {}
    for rf, ls in {(int(a), int(b)) for a, b in
                   zip(g["rflag"][mg].tolist(), g["lstatus"][mg].tolist())}:
        sel = mg & (g["rflag"] == rf) & (g["lstatus"] == ls)
        exp_g[(rf, ls)] = [int(np.count_nonzero(sel)),
                           int(g["qty"][sel].sum())]
    assert len(grouped) == len(exp_g)
    for k0, k1, vals in grouped:
        assert vals == exp_g[(k0, k1)]
    # baked-kernel repeat through the cross-segment combine
    assert results[0]["plan_grouped_repeat"] == grouped


def test_world2_q1_q3_q5_bitexact(tmp_path):
    results = run_world(2, 1, tmp_path)
    check_results(results, 2, 1)


def test_world4_q1_q3_q5_bitexact(tmp_path):
    results = run_world(4, 1, tmp_path)
    check_results(results, 4, 1)
