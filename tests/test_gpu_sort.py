"""GPU LSB radix sort (the general ORDER BY operator, nodeSort.c:48 /
tuplesort.c semantics) vs numpy's stable sort."""
import numpy as np
import pytest

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def eng():
    from greengage_amd import Engine
    e = Engine(device=0, n_segments=1, segment_id=0)
    yield e
    e.shutdown()


def test_radix_sort_keys_ascending(eng):
    rng = np.random.default_rng(7)
    for n in (0, 1, 63, 64, 65, 4096, 4097, 1_000_000):
        keys = rng.integers(0, 2**63, size=n, dtype=np.uint64)
        got, _ = eng.radix_sort(keys.copy())
        assert np.array_equal(got, np.sort(keys)), n


def test_radix_sort_stable_with_payload(eng):
    rng = np.random.default_rng(8)
    n = 500_000
    keys = rng.integers(0, 1000, size=n, dtype=np.uint64)  # many ties
    pay = np.arange(n, dtype=np.uint64)
    gk, gp = eng.radix_sort(keys.copy(), pay.copy(), key_bytes=2)
    order = np.argsort(keys, kind="stable")
    assert np.array_equal(gk, keys[order])
    assert np.array_equal(gp, order.astype(np.uint64))  # stability


def test_radix_sort_descending(eng):
    rng = np.random.default_rng(9)
    n = 200_000
    keys = rng.integers(0, 2**32, size=n, dtype=np.uint64)
    pay = np.arange(n, dtype=np.uint64)
    gk, gp = eng.radix_sort(keys.copy(), pay.copy(), key_bytes=4,
                            descending=True)
    order = np.argsort(-keys.astype(np.int64), kind="stable")
    assert np.array_equal(gk, keys[order])
    assert np.array_equal(gp, order.astype(np.uint64))


def test_radix_sort_partial_key_bytes(eng):
    """key_bytes=1 sorts only by the low byte (stable elsewhere) —
    the multi-key column-at-a-time building block (tuplesort_mk)."""
    keys = np.array([0x201, 0x100, 0x202, 0x101], dtype=np.uint64)
    gk, _ = eng.radix_sort(keys.copy(), key_bytes=1)
    assert gk.tolist() == [0x100, 0x201, 0x101, 0x202]
