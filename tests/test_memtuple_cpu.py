"""MemTuple binding parity (no GPU needed): the engine's restated
layout (gg_engine_memtuple_binding, host-only) must match the
reference's own create_memtuple_binding (memtuple.c compiled in place)
attribute for attribute — offsets, lengths, aligned lengths, null
byte/mask — across schema shapes."""
import numpy as np
import pytest

import pyoracle

ALIGN_OF = {8: "d", 4: "i", 2: "s", 1: "c"}

SCHEMAS = [
    [8, 4, 4],                   # Q3 motion tuple (okey, odate, prio)
    [8],
    [4],
    [4, 4],
    [2, 1, 8, 4],                # mixed, needs reordering
    [1, 1, 1],
    [2, 2, 2, 2],
    [8, 8, 8, 8, 8, 8, 8, 8],
    [4, 8, 1, 2, 8, 4, 1, 2, 4],
    [1] * 12,                    # bitmap > 1 byte, col_align 4
    [8] + [1] * 11,              # bitmap > 4 avail bytes, col_align 8
    [4] * 9,                     # 9 attrs, col_align 4
]


def _schema(attlen):
    return attlen, [ALIGN_OF[l] for l in attlen]


def _engine_binding(attlen, attalign):
    import sys
    sys.path.insert(0, ".")
    from greengage_amd.engine import Engine
    return Engine.memtuple_binding(attlen, attalign)


def test_binding_matches_reference():
    if pyoracle.dsb_ref() is None:
        pytest.skip("reference memtuple codec not built")
    for attlen in SCHEMAS:
        attlen, attalign = _schema(attlen)
        ref = pyoracle.MtSchema(attlen, attalign)
        (calign, extra, var_start), per = _engine_binding(attlen,
                                                          attalign)
        rcalign, rextra, rvar = ref.info()
        assert (calign, extra, var_start) == (rcalign, rextra, rvar), \
            attlen
        for a in range(1, len(attlen) + 1):
            roff, rlen, rlen_al, rflag, rnb, rnm = ref.colbind(a)
            eoff, elen, elen_al, enb, enm = per[a - 1]
            assert (eoff, elen, elen_al, enb, enm) == \
                (roff, rlen, rlen_al, rnb, rnm), (attlen, a)
            # large binding is identical for fixed-width attrs
            loff, llen, llen_al, lflag, lnb, lnm = ref.colbind(
                a, large=True)
            assert (loff, llen, llen_al, lnb, lnm) == \
                (roff, rlen, rlen_al, rnb, rnm), (attlen, a)


def test_reference_form_getattr_roundtrip():
    """Sanity on the wrapper itself: reference form→getattr round
    trips values and nulls."""
    if pyoracle.dsb_ref() is None:
        pytest.skip("reference memtuple codec not built")
    rng = np.random.default_rng(41)
    for attlen in SCHEMAS:
        attlen, attalign = _schema(attlen)
        ref = pyoracle.MtSchema(attlen, attalign)
        lim = [2 ** (8 * min(l, 7) - 1) for l in attlen]
        for trial in range(8):
            vals = [int(rng.integers(-m, m)) for m in lim]
            isnull = [int(rng.random() < 0.3) for _ in attlen]
            if trial == 0:
                isnull = [0] * len(attlen)
            tup = ref.form(vals, isnull)
            hdr = int(np.frombuffer(tup[:4].tobytes(), np.uint32)[0])
            assert hdr & 0x80000000
            assert (hdr & 0x3FFFFFF8) == len(tup)
            assert (hdr & 1) == (1 if any(isnull) else 0)
            for a in range(1, len(attlen) + 1):
                v, isn = ref.getattr(tup, a)
                assert isn == isnull[a - 1], (attlen, trial, a)
                if not isn:
                    width = attlen[a - 1]
                    mask = (1 << (8 * width)) - 1
                    assert v & mask == vals[a - 1] & mask, \
                        (attlen, trial, a)


# ---------------- varlena (text) attrs ----------------

TEXT_SCHEMAS = [
    ([8, -1, 4], "dii"),          # int8, text, int4
    ([-1], "i"),
    ([-1, -1, -1], "iii"),
    ([4, -1, 2, -1, 8], "disid"),  # mixed, reordering + 2 texts
    ([1, -1, 1], "ici"),
]


def test_binding_matches_reference_text():
    """Binding parity for schemas with varlena attrs: text binds as a
    2-byte varoffset in the 's' pass (MTB_ByRef) — offsets, aligned
    lengths and null bits must match the reference exactly."""
    if pyoracle.dsb_ref() is None:
        pytest.skip("reference memtuple codec not built")
    for attlen, attalign in TEXT_SCHEMAS:
        ref = pyoracle.MtSchema(attlen, list(attalign))
        (calign, extra, var_start), per = _engine_binding(
            attlen, list(attalign))
        rcalign, rextra, rvar = ref.info()
        assert (calign, extra, var_start) == (rcalign, rextra, rvar), \
            attlen
        for a in range(1, len(attlen) + 1):
            roff, rlen, rlen_al, rflag, rnb, rnm = ref.colbind(a)
            eoff, elen, elen_al, enb, enm = per[a - 1]
            assert (eoff, elen, elen_al, enb, enm) == \
                (roff, rlen, rlen_al, rnb, rnm), (attlen, a)
            if attlen[a - 1] == -1:
                assert rflag == 3, (attlen, a)  # MTB_ByRef


def test_reference_form_var_header():
    """Wrapper sanity for mixed tuples: formed length/flags look right
    and short/4B text forms appear in the varlen section."""
    if pyoracle.dsb_ref() is None:
        pytest.skip("reference memtuple codec not built")
    ref = pyoracle.MtSchema([8, -1, 4], ["d", "i", "i"])
    tup = ref.form_var([7, b"hello", -3], [0, 0, 0])
    hdr = int(np.frombuffer(tup[:4].tobytes(), np.uint32)[0])
    assert hdr & 0x80000000 and (hdr & 0x3FFFFFF8) == len(tup)
    assert b"hello" in tup.tobytes()
    # long text -> 4-byte network-order header in the tail
    long = b"Q" * 300
    tup2 = ref.form_var([7, long, -3], [0, 0, 0])
    assert long in tup2.tobytes()
    assert len(tup2) > 300


def test_binding_matches_reference_large():
    """LARGE binding (4-byte varoffsets, used over 0xFFF0 bytes): the
    engine's large layout must equal the reference's large_bind."""
    if pyoracle.dsb_ref() is None:
        pytest.skip("reference memtuple codec not built")
    import sys
    sys.path.insert(0, ".")
    from greengage_amd.engine import Engine
    for attlen, attalign in TEXT_SCHEMAS:
        ref = pyoracle.MtSchema(attlen, list(attalign))
        meta, per = Engine.memtuple_binding_large(attlen, list(attalign))
        for a in range(1, len(attlen) + 1):
            roff, rlen, rlen_al, rflag, rnb, rnm = ref.colbind(
                a, large=True)
            eoff, elen, elen_al, enb, enm = per[a - 1]
            assert (eoff, elen, elen_al, enb, enm) == \
                (roff, rlen, rlen_al, rnb, rnm), (attlen, a)
            if attlen[a - 1] == -1:
                assert elen == 4, (attlen, a)
