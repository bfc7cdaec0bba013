"""The reference's own AOCS datum-stream codec (compiled in place,
oracle/_ref/libpg_dsbref.so) round-trips — validates the wrapper before
it is used as the GPU decoder's parity oracle."""
import numpy as np
import pytest

import pyoracle


def patterns(rng, n, datumlen):
    lim = 2**31 - 1 if datumlen == 4 else 2**62
    return {
        "random": (rng.integers(-lim, lim, n).astype(np.int64),
                   (rng.random(n) < 0.1).astype(np.uint8)),
        "rle": (np.repeat(rng.integers(0, 50, max(n // 100, 1)), 100)[:n]
                .astype(np.int64), np.zeros(n, np.uint8)),
        "delta": ((np.arange(n) * 3 + rng.integers(0, 3, n)).astype(np.int64),
                  np.zeros(n, np.uint8)),
        "nulls_heavy": (rng.integers(-lim, lim, n).astype(np.int64),
                        (rng.random(n) < 0.7).astype(np.uint8)),
        "all_null": (np.zeros(n, np.int64), np.ones(n, np.uint8)),
        "single": (np.array([42] * n, np.int64), np.zeros(n, np.uint8)),
    }


CFGS = [(0, 0, 0), (1, 0, 0), (1, 1, 0), (2, 1, 0), (2, 1, 1)]


def test_reference_codec_roundtrip():
    if pyoracle.dsb_ref() is None:
        pytest.skip("reference dsb codec not built")
    rng = np.random.default_rng(13)
    for datumlen in (4, 8):
        for name, (vals, nulls) in patterns(rng, 20000, datumlen).items():
            if datumlen == 4:
                vals = vals.astype(np.int32).astype(np.int64)
            for version, rle, delta in CFGS:
                stream, nb = pyoracle.dsb_encode(vals, nulls, datumlen,
                                                 version, rle, delta)
                dv, dn = pyoracle.dsb_decode(stream, datumlen, version,
                                             rle, len(vals) + 10)
                assert len(dv) == len(vals), (name, version)
                assert np.array_equal(dn != 0, nulls != 0), (name, version)
                mask = nulls == 0
                assert np.array_equal(dv[mask], vals[mask]), (name, version)
