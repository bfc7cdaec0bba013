"""GPU MemTuple codec vs the REFERENCE's own memtuple.c (compiled in
place): bulk-encoded streams must be byte-identical to reference-formed
tuples, and decode must round-trip both engine- and reference-produced
streams."""
import numpy as np
import pytest

import pyoracle
from test_memtuple_cpu import SCHEMAS, _schema

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def eng():
    from greengage_amd import Engine
    e = Engine(device=0, n_segments=1, segment_id=0)
    yield e
    e.shutdown()


DT = {1: np.uint8, 2: np.int16, 4: np.int32, 8: np.int64}


def _random_cols(rng, attlen, nrows, null_frac):
    cols, nulls = [], []
    for l in attlen:
        lim = 2 ** (8 * min(l, 7) - 1)
        cols.append(rng.integers(-lim if l > 1 else 0, lim,
                                 nrows).astype(DT[l]))
        nulls.append((rng.random(nrows) < null_frac).astype(np.uint8)
                     if null_frac else None)
    return cols, nulls


def test_gpu_encode_matches_reference_bytes(eng):
    if pyoracle.dsb_ref() is None:
        pytest.skip("reference memtuple codec not built")
    from greengage_amd.engine import Engine as E
    rng = np.random.default_rng(42)
    nrows = 500
    for attlen in SCHEMAS:
        attlen, attalign = _schema(attlen)
        ref = pyoracle.MtSchema(attlen, attalign)
        for null_frac in (0.0, 0.3):
            cols, nulls = _random_cols(rng, attlen, nrows, null_frac)
            stream = E.memtuple_encode(attlen, attalign, cols, nulls)
            # reference forms each tuple; concatenation must be
            # byte-identical
            refbytes = []
            for r in range(nrows):
                vals = [int(c[r]) for c in cols]
                isnull = [0 if n is None else int(n[r]) for n in nulls]
                refbytes.append(ref.form(vals, isnull))
            refstream = np.concatenate(refbytes)
            assert len(stream) == len(refstream), (attlen, null_frac)
            assert np.array_equal(stream, refstream), \
                (attlen, null_frac)


def test_gpu_decode_roundtrip(eng):
    if pyoracle.dsb_ref() is None:
        pytest.skip("reference memtuple codec not built")
    from greengage_amd.engine import Engine as E
    rng = np.random.default_rng(43)
    nrows = 2000
    for attlen in SCHEMAS[:6]:
        attlen, attalign = _schema(attlen)
        cols, nulls = _random_cols(rng, attlen, nrows, 0.25)
        stream = E.memtuple_encode(attlen, attalign, cols, nulls)
        dcols, dnulls = E.memtuple_decode(attlen, attalign, stream,
                                          nrows + 10)
        for i, l in enumerate(attlen):
            assert np.array_equal(dnulls[i] != 0, nulls[i] != 0), \
                (attlen, i)
            mask = nulls[i] == 0
            assert np.array_equal(dcols[i][mask], cols[i][mask]), \
                (attlen, i)


def test_gpu_decode_reference_stream(eng):
    """Streams formed by the REFERENCE decode correctly on the GPU."""
    if pyoracle.dsb_ref() is None:
        pytest.skip("reference memtuple codec not built")
    from greengage_amd.engine import Engine as E
    rng = np.random.default_rng(44)
    attlen, attalign = _schema([8, 4, 4, 2, 1])
    ref = pyoracle.MtSchema(attlen, attalign)
    nrows = 300
    cols, nulls = _random_cols(rng, attlen, nrows, 0.4)
    parts = []
    for r in range(nrows):
        vals = [int(c[r]) for c in cols]
        isnull = [int(n[r]) for n in nulls]
        parts.append(ref.form(vals, isnull))
    stream = np.concatenate(parts)
    dcols, dnulls = E.memtuple_decode(attlen, attalign, stream,
                                      nrows + 10)
    assert len(dcols[0]) == nrows
    for i in range(len(attlen)):
        assert np.array_equal(dnulls[i] != 0, nulls[i] != 0), i
        mask = nulls[i] == 0
        assert np.array_equal(dcols[i][mask], cols[i][mask]), i


def test_gpu_encode_text_matches_reference_bytes(eng):
    """Mixed fixed/text tuples: GPU bulk encode must be byte-identical
    to the reference's memtuple_form_to with text datums (short and
    4-byte forms, nulls)."""
    if pyoracle.dsb_ref() is None:
        pytest.skip("reference memtuple codec not built")
    from greengage_amd.engine import Engine as E
    from test_memtuple_cpu import TEXT_SCHEMAS
    rng = np.random.default_rng(45)
    nrows = 400
    for attlen, attalign in TEXT_SCHEMAS:
        attalign = list(attalign)
        ref = pyoracle.MtSchema(attlen, attalign)
        for null_frac in (0.0, 0.3):
            cols, nulls = [], []
            for l in attlen:
                if l == -1:
                    vals = []
                    for r in range(nrows):
                        ln = int(rng.integers(0, 40)) if \
                            rng.random() < 0.9 else \
                            int(rng.integers(120, 400))
                        vals.append(bytes(rng.integers(65, 91, ln)
                                          .astype(np.uint8)))
                    cols.append(vals)
                else:
                    lim = 2 ** (8 * min(l, 7) - 1)
                    cols.append(rng.integers(
                        -lim if l > 1 else 0, lim, nrows).astype(DT[l]))
                nulls.append((rng.random(nrows) < null_frac)
                             .astype(np.uint8) if null_frac else None)
            stream = E.memtuple_encode(attlen, attalign, cols, nulls)
            parts = []
            for r in range(nrows):
                vals = [c[r] if attlen[i] == -1 else int(c[r])
                        for i, c in enumerate(cols)]
                isnull = [0 if n is None else int(n[r]) for n in nulls]
                parts.append(ref.form_var(vals, isnull))
            refstream = np.concatenate(parts)
            assert len(stream) == len(refstream), (attlen, null_frac)
            assert np.array_equal(stream, refstream), \
                (attlen, null_frac)
            # and the decode round-trips
            dcols, dnulls = E.memtuple_decode(attlen, attalign, stream,
                                              nrows + 10)
            for i, l in enumerate(attlen):
                if nulls[i] is not None:
                    assert np.array_equal(dnulls[i] != 0,
                                          nulls[i] != 0), (attlen, i)
                for r in range(nrows):
                    if nulls[i] is not None and nulls[i][r]:
                        continue
                    if l == -1:
                        assert dcols[i][r] == cols[i][r], (attlen, i, r)
                    else:
                        assert dcols[i][r] == cols[i][r], (attlen, i, r)


def test_gpu_encode_large_tuples(eng):
    """Tuples over MEMTUPLE_LEN_FITSHORT switch to the LARGE binding
    (4-byte varoffsets, MEMTUP_LARGETUP flag) — byte-identical to the
    reference, mixed with small tuples in one stream."""
    if pyoracle.dsb_ref() is None:
        pytest.skip("reference memtuple codec not built")
    from greengage_amd.engine import Engine as E
    rng = np.random.default_rng(46)
    attlen, attalign = [8, -1, -1, 4], ["d", "i", "i", "i"]
    ref = pyoracle.MtSchema(attlen, attalign)
    nrows = 60
    t1, t2, ints1, ints4 = [], [], [], []
    for r in range(nrows):
        if r % 3 == 0:  # large row: two ~40KB payloads
            t1.append(bytes(rng.integers(65, 91, 40000).astype(np.uint8)))
            t2.append(bytes(rng.integers(97, 123, 30000).astype(np.uint8)))
        else:
            t1.append(b"small-" + bytes([65 + r % 26]))
            t2.append(bytes(rng.integers(65, 91, int(rng.integers(0, 200)))
                            .astype(np.uint8)))
        ints1.append(r * 1000)
        ints4.append(-r)
    cols = [np.array(ints1, np.int64), t1, t2, np.array(ints4, np.int32)]
    nulls = [(rng.random(nrows) < 0.2).astype(np.uint8)
             for _ in range(4)]
    stream = E.memtuple_encode(attlen, attalign, cols, nulls)
    parts = []
    for r in range(nrows):
        vals = [int(cols[0][r]), t1[r], t2[r], int(cols[3][r])]
        isnull = [int(n[r]) for n in nulls]
        parts.append(ref.form_var(vals, isnull))
    refstream = np.concatenate(parts)
    assert len(stream) == len(refstream)
    assert np.array_equal(stream, refstream)
    # large flag present on the big rows
    pos = 0
    nlarge = 0
    while pos < len(stream):
        hdr = int(stream[pos:pos + 4].view(np.uint32)[0])
        if hdr & 2:
            nlarge += 1
        pos += hdr & 0x3FFFFFF8
    assert nlarge > 0
    # round trip
    dcols, dnulls = E.memtuple_decode(attlen, attalign, stream,
                                      nrows + 5)
    for i, l in enumerate(attlen):
        assert np.array_equal(dnulls[i] != 0, nulls[i] != 0), i
        for r in range(nrows):
            if nulls[i][r]:
                continue
            assert dcols[i][r] == (cols[i][r] if l == -1
                                   else cols[i][r]), (i, r)
