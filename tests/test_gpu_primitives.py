"""Device primitive isolation tests: radix sort and hash key->slot at high
cardinalities (these back every keyed GPU operator)."""
import numpy as np
import pytest

from windflow_amd import _core

pytestmark = pytest.mark.gpu


@pytest.mark.parametrize("n,bits", [(100_000, 11), (1_000_000, 21),
                                    (4_000_000, 21), (1_000_003, 14)])
def test_sort_pairs_matches_numpy_stable(n, bits):
    rng = np.random.default_rng(7)
    keys = rng.integers(0, 1 << bits, size=n, dtype=np.uint32)
    ks, vs = _core.debug_sort_pairs(keys, bits)
    order = np.argsort(keys, kind="stable")
    assert np.array_equal(ks, keys[order])
    assert np.array_equal(vs, order.astype(np.uint32))


def test_key_to_slot_bijective_1m():
    rng = np.random.default_rng(11)
    keys = rng.integers(0, 1_000_000, size=4_000_000, dtype=np.uint64)
    slots = _core.debug_key_slots(keys, 2_000_000)
    # same key -> same slot; distinct keys -> distinct slots
    first = {}
    k2s = {}
    for k, s in zip(keys.tolist(), slots.tolist()):
        if k in k2s:
            assert k2s[k] == s
        else:
            k2s[k] = s
    assert len(set(k2s.values())) == len(k2s)
    assert max(k2s.values()) == len(k2s) - 1
