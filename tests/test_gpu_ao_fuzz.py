"""Seeded AO-decode fuzz: random columns ENCODED BY THE REFERENCE'S
OWN WRITER (oracle/ref_build compiles datumstreamblock.c +
cdbappendonlystorageformat.c in place), mounted through
gg_engine_register_table_ao and DECODED BY THE ENGINE'S GPU kernels
(aocs_decode.hip), validated value-exactly via COUNT/COUNT(col)/SUM
plans against numpy.

Grid per iteration: datum width (4/8) x DatumStream version/rle/delta
combination x nullability x value shape (random / RLE runs /
monotonic delta-friendly) x AO framing (plain, zlib, zstd), with row
counts crossing block boundaries.  Cites: datumstreamblock.c (write
paths), cdbappendonlystorageformat.c (block headers/CRC).
"""
import numpy as np
import pytest

import pyoracle

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def eng():
    import torch
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from greengage_amd import Engine
    e = Engine()
    yield e
    e.shutdown()


# (dsb_version, rle, delta) combos the reference writer supports
DSB_MODES = [(1, 0, 0), (1, 1, 0), (2, 0, 0), (2, 1, 0), (2, 1, 1)]
NEG_INF = -(1 << 63)
POS_INF = (1 << 63) - 1


def _values(rng, n, width, shape):
    lim = 1000 if width == 4 else 100_000
    if shape == 0:          # random
        v = rng.integers(-lim, lim, n)
    elif shape == 1:        # RLE-friendly runs
        v = rng.integers(-lim, lim, n)
        v[rng.random(n) < 0.6] = int(rng.integers(-5, 5))
    else:                   # monotonic-ish (delta-friendly)
        v = np.cumsum(rng.integers(0, 4, n)) - lim // 2
    return v.astype(np.int64)


def test_ao_decode_fuzz(eng):
    if pyoracle.dsb_ref() is None:
        pytest.skip("reference dsb codec not built")
    rng = np.random.default_rng(20260916)
    it = 0
    for width in (4, 8):
        for (ver, rle, delta) in DSB_MODES:
            n = int(rng.integers(1_000, 180_000))
            shape = int(rng.integers(0, 3))
            nullable = bool(rng.random() < 0.5)
            framing = int(rng.integers(0, 3))

            vals = _values(rng, n, width, shape)
            nulls = ((rng.random(n) < 0.15) if nullable
                     else np.zeros(n, bool)).astype(np.uint8)
            framed, _nb = pyoracle.dsb_encode(
                vals, nulls, width, ver, rle, delta)
            if framing == 0:
                ao, ct = pyoracle.ao_wrap(framed), 0
            elif framing == 1:
                ao, ct = pyoracle.ao_wrap_compressed(framed, 1, 6), 1
            else:
                ao, ct = pyoracle.ao_wrap_compressed(framed, 2, 3), 2

            t = eng.register_table_ao(f"aofz{it}", [
                ("v", "int32" if width == 4 else "int64", ao, 1, 2,
                 ver, ct, 0, 1 if nullable else 0)])
            # no predicate: COUNT(*) counts every row, COUNT(v)/SUM(v)
            # are strict over the null flags -> validates both the
            # decoded values and the null bitmap
            p = eng.compile_plan(t, aggs=[
                "count", ("count", "v"), ("sum", [("v", "id")])])
            g = eng.execute_plan(p, max_groups=8)
            nn = nulls == 0
            label = (f"it={it} w={width} v={ver} rle={rle} "
                     f"delta={delta} shape={shape} null={nullable} "
                     f"frame={framing} n={n}")
            assert g[0][2][0] == n, label
            assert g[0][2][1] == int(np.count_nonzero(nn)), label
            assert g[0][2][2] == int(vals[nn].sum()), label

            # bounded predicate leg: NULL fails the qual
            lo = int(rng.integers(-50, 0))
            hi = int(rng.integers(1, 60))
            p2 = eng.compile_plan(t, preds=[("v", lo, hi)],
                                  aggs=["count",
                                        ("sum", [("v", "id")])])
            g2 = eng.execute_plan(p2, max_groups=8)
            m = nn & (vals >= lo) & (vals < hi)
            assert g2[0][2][0] == int(np.count_nonzero(m)), label
            assert g2[0][2][1] == int(vals[m].sum()), label
            it += 1
