"""Pin the restated hashing (include/gg_pg_hash.h, via liboracle.so)
against the reference's own hashfunc.c:

  - committed golden vectors (tests/golden/hash_vectors.json) produced by
    the reference library compiled in place (oracle/ref_build/)
  - a live fuzz comparison against that library when it is present
"""
import random

import pytest

import pyoracle


def test_hash_any_vectors(golden):
    L = pyoracle.lib()
    for hexstr, expect in golden("hash_vectors.json")["hash_any"]:
        b = bytes.fromhex(hexstr)
        assert L.gg_oracle_hash_any(b, len(b)) == expect


def test_scalar_hash_vectors(golden):
    L = pyoracle.lib()
    v = golden("hash_vectors.json")
    for x, expect in v["hashint4"]:
        assert L.gg_oracle_hashint4(x) == expect
    for x, expect in v["hashint8"]:
        assert L.gg_oracle_hashint8(x) == expect
    for x, expect in v["hash_uint32"]:
        assert L.gg_oracle_hash_uint32(x) == expect
    for c, expect in v["hashchar"]:
        assert L.gg_oracle_hashchar(c.encode()) == expect


def test_hash_any_live_fuzz_vs_reference():
    R = pyoracle.ref()
    if R is None:
        pytest.skip("reference library not present")
    L = pyoracle.lib()
    rng = random.Random(123)
    for _ in range(5000):
        n = rng.randrange(0, 64)
        b = bytes(rng.randrange(256) for _ in range(n))
        assert L.gg_oracle_hash_any(b, n) == R.ref_hash_any(b, n)
    for _ in range(2000):
        v = rng.randrange(-2**63, 2**63)
        assert L.gg_oracle_hashint8(v) == R.ref_hashint8(v)
        w = rng.randrange(-2**31, 2**31)
        assert L.gg_oracle_hashint4(w) == R.ref_hashint4(w)


def test_jump_consistent_hash_properties():
    """jump_consistent_hash (cdbhash.c:549, arXiv:1406.2294):
    range [0, n); consistency: growing n only moves keys INTO the new
    bucket, never between old buckets."""
    L = pyoracle.lib()
    rng = random.Random(7)
    keys = [rng.randrange(0, 2**64) for _ in range(2000)]
    for n in (1, 2, 3, 8):
        for k in keys[:200]:
            s = L.gg_oracle_jump_hash(k, n)
            assert 0 <= s < n
    for k in keys:
        prev = L.gg_oracle_jump_hash(k, 7)
        nxt = L.gg_oracle_jump_hash(k, 8)
        assert nxt == prev or nxt == 7


def test_segment_mapping_balance():
    """cdbhash(l_orderkey) → segment spreads dense orderkeys evenly."""
    L = pyoracle.lib()
    counts = [0] * 8
    for key in range(1, 20001):
        counts[L.gg_oracle_segment_int8(key, 8)] += 1
    assert min(counts) > 0.8 * (20000 / 8)
    assert max(counts) < 1.2 * (20000 / 8)


def test_segment_int4_int8_consistent():
    """hashint8 of a small positive value equals hashint4 of it
    (hashfunc.c:52 comment: cross-type hash-join compatibility) — so
    integer o_custkey and bigint keys land on the same segment."""
    L = pyoracle.lib()
    for v in (1, 2, 12345, 2**31 - 1):
        assert L.gg_oracle_hashint4(v) == L.gg_oracle_hashint8(v)
        assert L.gg_oracle_segment_int4(v, 8) == L.gg_oracle_segment_int8(v, 8)
