"""Seeded random-descriptor fuzz: the generic plan path vs a python
restatement of the reference executor's semantics.

Each iteration builds a random gg_plan_desc — random column types
(int64/int32/dec64/char1), random nullability, 0..3 scan range
conjuncts (ExecQual over half-open ranges, one-sided via
NEG_INF/POS_INF), 0..2 hash SEMI-joins (nodeHash.c build +
nodeHashjoin.c probe; NULL keys never match, nodeHash.c:1070), group
by 0/1/2 columns (NULL keys grouped distinct, execHHashagg.c:531),
and 1..4 aggregates (COUNT(*) / COUNT(col) / SUM of factor products
with strict transitions, nodeAgg.c:413) — then checks the GPU result
(hipRTC-specialized kernel) against exact big-int python evaluation:
same group set, same order (sorted, NULL first), bit-identical
128-bit sums.  Key ranges are drawn to exercise both the dense-bitmap
and the open-addressing hash build paths.
"""
import os

import numpy as np
import pytest

pytestmark = pytest.mark.gpu

NULL_KEY = -(1 << 63) + 1
NEG_INF = -(1 << 63)
POS_INF = (1 << 63) - 1


@pytest.fixture(scope="module")
def eng():
    import torch
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from greengage_amd import Engine
    e = Engine()
    yield e
    e.shutdown()


TYPES = [("int64", np.int64), ("int32", np.int32),
         ("dec64", np.int64), ("char1", np.uint8)]


def _mk_table(eng, rng, name, nrows, want_char1=0):
    """Random table: 1 join-key col, 2-4 value cols, optional char1
    group cols, each col independently nullable."""
    cols, data, nulls = [], {}, {}

    # join key: dense-ish (dense bitmap path) or sparse/negative (hash)
    style = rng.integers(0, 3)
    if style == 0:
        k = rng.integers(0, max(1, nrows // 2), nrows)
    elif style == 1:
        k = rng.integers(0, nrows * 64, nrows)       # > 8x rows: hash
    else:
        k = rng.integers(-1000, 1000, nrows)          # negative: hash
    cols.append(("k", "int64", k.astype(np.int64)))

    nval = int(rng.integers(2, 5))
    for i in range(nval):
        tname, npty = TYPES[int(rng.integers(0, len(TYPES)))]
        if tname == "char1":
            a = rng.integers(0, 256, nrows).astype(np.uint8)
        elif tname == "int32":
            a = rng.integers(-5000, 5000, nrows).astype(np.int32)
        else:
            a = rng.integers(-120, 120, nrows).astype(np.int64)
        cols.append((f"v{i}", tname, a))
    for i in range(want_char1):
        # small random alphabet: 2x2 .. 6x6 group grids, so repeated
        # executes sometimes cross the <=8-group bake threshold and
        # sometimes stay on the generic grouped kernel
        a = rng.integers(0, int(rng.integers(2, 7)),
                         nrows).astype(np.uint8)
        cols.append((f"g{i}", "char1", a))

    t = eng.register_table(name, cols, nrows)
    for cname, tname, a in cols:
        data[cname] = (a.astype(object), tname)
        if rng.random() < 0.35:
            nl = (rng.random(nrows) < rng.uniform(0.02, 0.3)) \
                .astype(np.uint8)
            eng.set_nulls(t, cname, nl)
            nulls[cname] = nl.astype(bool)
        else:
            nulls[cname] = np.zeros(nrows, bool)
    return t, cols, data, nulls


def _rand_preds(rng, cols, nulls_unused, nmax):
    preds = []
    for _ in range(int(rng.integers(0, nmax + 1))):
        cname, tname, a = cols[int(rng.integers(0, len(cols)))]
        vals = a.astype(np.int64)
        lo_v, hi_v = int(vals.min()), int(vals.max()) + 1
        # random sub-range, sometimes one-sided / empty / full
        p = rng.random()
        if p < 0.1:
            lo, hi = NEG_INF, int(rng.integers(lo_v, hi_v + 1))
        elif p < 0.2:
            lo, hi = int(rng.integers(lo_v, hi_v + 1)), POS_INF
        elif p < 0.25:
            lo, hi = hi_v + 5, hi_v + 6   # selects nothing
        else:
            a1 = int(rng.integers(lo_v, hi_v + 1))
            b1 = int(rng.integers(lo_v, hi_v + 1))
            lo, hi = min(a1, b1), max(a1, b1)
        preds.append((cname, lo, hi))
    return preds


def _pred_mask(preds, data, nulls, n):
    m = np.ones(n, bool)
    for cname, lo, hi in preds:
        v = data[cname][0].astype(np.int64)
        m &= ~nulls[cname] & (v >= lo) & (v < hi)
    return m


def _eval_plan(n, data, nulls, preds, joins, group_cols, aggs):
    """Exact big-int restatement of the descriptor semantics."""
    m = _pred_mask(preds, data, nulls, n)
    for j in joins:
        bn, bdata, bnulls = j["_n"], j["_data"], j["_nulls"]
        bm = _pred_mask(j.get("preds", ()), bdata, bnulls, bn)
        bk = j["build_key"]
        keyset = set(
            bdata[bk][0][bm & ~bnulls[bk]].astype(np.int64).tolist())
        pk = j["probe_key"]
        pv = data[pk][0].astype(np.int64)
        hit = np.fromiter((int(x) in keyset for x in pv), bool, n)
        m &= ~nulls[pk] & hit
    idx = np.nonzero(m)[0]

    def gkey(i):
        out = []
        for g, c in enumerate(group_cols):
            out.append(NULL_KEY if nulls[c][i]
                       else int(data[c][0][i]))
        while len(out) < 2:
            out.append(0)
        return tuple(out)

    groups = {}
    if not group_cols:
        # plain aggregate: exactly one output row even over zero
        # input rows (nodeAgg emits the initial transvalues; the
        # arena represents empty SUMs as 0 with COUNT=0 available to
        # the finalizer to distinguish NULL)
        groups[(0, 0)] = [0] * len(aggs)
    for i in idx.tolist():
        key = gkey(i)
        acc = groups.setdefault(key, [0] * len(aggs))
        for a, spec in enumerate(aggs):
            if spec == "count":
                acc[a] += 1
                continue
            if spec[0] == "count":
                if not nulls[spec[1]][i]:
                    acc[a] += 1
                continue
            factors = spec[1]
            if any(nulls[c][i] for c, _m in factors):
                continue
            v = 1
            for c, mod in factors:
                x = int(data[c][0][i])
                if mod == "sub100":
                    x = 100 - x
                elif mod == "add100":
                    x = 100 + x
                v *= x
            acc[a] += v
    return sorted((k[0], k[1], vals) for k, vals in groups.items())


def _rand_aggs(rng, cols):
    aggs = []
    numeric = [c for c in cols if c[1] != "char1"]
    for _ in range(int(rng.integers(1, 5))):
        p = rng.random()
        if p < 0.25:
            aggs.append("count")
        elif p < 0.45:
            aggs.append(("count",
                         numeric[int(rng.integers(0, len(numeric)))][0]))
        else:
            nf = int(rng.integers(1, 4))
            fs = []
            for _ in range(nf):
                c = numeric[int(rng.integers(0, len(numeric)))][0]
                fs.append((c, ["id", "sub100", "add100"]
                           [int(rng.integers(0, 3))]))
            aggs.append(("sum", fs))
    return aggs


def _run_one(eng, rng, it, force_interp=False):
    n = int(rng.integers(1, 300_000))
    t, cols, data, nulls = _mk_table(
        eng, rng, f"fz{it}_d", n, want_char1=2)

    preds = _rand_preds(rng, cols[:-2], None, 3)
    joins = []
    for j in range(int(rng.integers(0, 3))):
        bn = int(rng.integers(1, 80_000))
        bt, bcols, bdata, bnulls = _mk_table(
            eng, rng, f"fz{it}_b{j}", bn)
        jspec = {"table": bt, "build_key": "k", "probe_key": "k",
                 "preds": _rand_preds(rng, bcols, None, 2),
                 "_n": bn, "_data": bdata, "_nulls": bnulls}
        joins.append(jspec)

    gm = int(rng.integers(0, 3))
    # group-1 key: a value column (bounded cardinality — the generic
    # group table holds 2^18 slots; the raw join key can have ~n
    # distinct values)
    group_cols = [] if gm == 0 else (
        [cols[int(rng.integers(1, len(cols) - 2))][0]]
        if gm == 1 else ["g0", "g1"])
    aggs = _rand_aggs(rng, cols)

    if force_interp:
        os.environ["GG_PLAN_RTC"] = "0"
    try:
        p = eng.compile_plan(
            t, preds=preds,
            joins=[{k: v for k, v in j.items()
                    if not k.startswith("_")} for j in joins],
            group_cols=group_cols, aggs=aggs)
        got = eng.execute_plan(p, max_groups=1 << 17)
    finally:
        os.environ.pop("GG_PLAN_RTC", None)

    exp = _eval_plan(n, data, nulls, preds, joins, group_cols, aggs)
    got_t = sorted((k0, k1, vals) for k0, k1, vals in got)
    assert len(got_t) == len(exp), \
        f"it={it}: {len(got_t)} groups vs {len(exp)}"
    for g, e in zip(got_t, exp):
        assert g[0] == e[0] and g[1] == e[1], f"it={it}: key {g} vs {e}"
        assert g[2] == e[2], f"it={it}: vals {g} vs {e}"

    # second execute: plans with <=8 observed groups re-run on the
    # baked-codes kernel — results must be identical
    got2 = sorted(eng.execute_plan(p, max_groups=1 << 17))
    assert got2 == got_t, f"it={it}: repeat/baked execute diverged"


def test_plan_fuzz_rtc(eng):
    rng = np.random.default_rng(20260915)
    for it in range(32):
        _run_one(eng, rng, it)


def test_plan_fuzz_interpreted(eng):
    rng = np.random.default_rng(777)
    for it in range(10):
        _run_one(eng, rng, 100 + it, force_interp=True)


def test_baked_group_codes_kernel(eng):
    """Q1-shaped repeat execution: after the first execute the 2x2
    char1 group set is baked into a direct-indexed RTC kernel
    (path_plan_rtc_baked stat row); executes 2..4 must stay
    bit-identical to the exact python evaluation."""
    rng = np.random.default_rng(99)
    n = 400_000
    g0 = rng.choice(np.frombuffer(b"AN", np.uint8), n)
    g1 = rng.choice(np.frombuffer(b"FO", np.uint8), n)
    q = rng.integers(1, 51, n).astype(np.int64)
    pr = rng.integers(900, 105000, n).astype(np.int64)
    d = rng.integers(0, 11, n).astype(np.int64)
    sd = rng.integers(0, 2526, n).astype(np.int64)
    t = eng.register_table("bake_q1", [
        ("g0", "char1", g0), ("g1", "char1", g1),
        ("q", "dec64", q), ("pr", "dec64", pr), ("d", "dec64", d),
        ("sd", "int64", sd)], n)
    data = {c: (a.astype(object), ty) for (c, ty, a) in [
        ("g0", "char1", g0), ("g1", "char1", g1), ("q", "dec64", q),
        ("pr", "dec64", pr), ("d", "dec64", d), ("sd", "int64", sd)]}
    nulls = {c: np.zeros(n, bool) for c in data}
    preds = [("sd", NEG_INF, 2400)]
    aggs = ["count", ("sum", [("q", "id")]), ("sum", [("pr", "id")]),
            ("sum", [("pr", "id"), ("d", "sub100")])]
    p = eng.compile_plan(t, preds=preds, group_cols=["g0", "g1"],
                         aggs=aggs)
    exp = _eval_plan(n, data, nulls, preds, [], ["g0", "g1"], aggs)
    runs = [sorted(eng.execute_plan(p)) for _ in range(4)]
    for r in runs:
        assert r == exp
    paths = {s["name"] for s in eng.stats(p)}
    assert paths & {"path_plan_rtc_baked", "path_plan_rtc_baked_fast",
                    "path_plan_interp"}  # interp-only boxes: no bake


def test_plan_fuzz_group1_with_nulls(eng):
    """Directed case: single int64 group key with NULLs — the
    GG_PLAN_NULL_KEY sentinel group must sort first and carry exact
    strict-transition sums."""
    rng = np.random.default_rng(5)
    n = 50_000
    k = rng.integers(0, 40, n).astype(np.int64)
    v = rng.integers(-120, 120, n).astype(np.int64)
    knull = (rng.random(n) < 0.1).astype(np.uint8)
    vnull = (rng.random(n) < 0.1).astype(np.uint8)
    t = eng.register_table("fz_nullgrp", [("k", "int64", k),
                                          ("v", "dec64", v)], n)
    eng.set_nulls(t, "k", knull)
    eng.set_nulls(t, "v", vnull)
    data = {"k": (k.astype(object), "int64"),
            "v": (v.astype(object), "dec64")}
    nulls = {"k": knull.astype(bool), "v": vnull.astype(bool)}
    aggs = ["count", ("count", "v"), ("sum", [("v", "sub100")])]
    p = eng.compile_plan(t, group_cols=["k"], aggs=aggs)
    got = sorted(eng.execute_plan(p))
    exp = _eval_plan(n, data, nulls, [], [], ["k"], aggs)
    assert got == exp
    assert got[0][0] == NULL_KEY     # NULL group present, first
