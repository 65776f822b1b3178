"""Pin the oracle's radix-sort restatement with the reference's own test
procedure: RadixSortSuite.scala:46-75 (sort-type table), :151-176 (seed-123
tests on XORShiftRandom data), :178-200 (bitmask fuzz), checked against an
INDEPENDENT sort (numpy stable argsort on an order-equivalent key transform)
rather than against the oracle itself."""
import numpy as np
import pytest

import oracle

N = 10000
SIGN = np.uint64(0x8000000000000000)

# (name, start_byte, end_byte, desc, signed) — RadixSortSuite.scala:46-75.
# nullsFirst is not part of the radix core (nulls never enter the radix array).
SORT_TYPES = [
    ("unsigned asc", 0, 7, False, False),
    ("unsigned desc", 0, 7, True, False),
    ("twos complement asc", 0, 7, False, True),
    ("twos complement desc", 0, 7, True, True),
    ("binary data partial", 2, 4, False, False),
]


def order_key(vals: np.ndarray, start_byte, end_byte, desc, signed) -> np.ndarray:
    """Map each value to a uint64 whose ascending unsigned order equals the
    reference comparator's order (PrefixComparators.java:104-190)."""
    u = vals.view(np.uint64).copy()
    lo, hi = 8 * start_byte, 8 * (end_byte + 1)
    mask = np.uint64((((1 << (hi - lo)) - 1) << lo) & 0xFFFFFFFFFFFFFFFF)
    u &= mask
    if signed:
        u ^= SIGN  # two's-complement order == unsigned order with sign flipped
    if desc:
        u = ~u
    return u


def expected_perm(vals, start_byte, end_byte, desc, signed):
    return np.argsort(order_key(vals, start_byte, end_byte, desc, signed), kind="stable")


@pytest.mark.parametrize("name,sb,eb,desc,signed", SORT_TYPES)
def test_sort_seed123(name, sb, eb, desc, signed):
    # RadixSortSuite.scala:151-160 "sort <type>": XORShiftRandom(123), N=10000
    data = oracle.XorShiftRandom(123).fill_longs(N)
    got = oracle.radix_sort_longs(data, sb, eb, desc, signed).view(np.int64)
    exp = data[expected_perm(data, sb, eb, desc, signed)]
    assert (got == exp).all()


@pytest.mark.parametrize("name,sb,eb,desc,signed", SORT_TYPES)
def test_sort_key_prefix_seed123(name, sb, eb, desc, signed):
    # RadixSortSuite.scala:162-173 "sort key prefix <type>": rand & 0xff pairs
    flat = oracle.XorShiftRandom(123).fill_longs(2 * N, mask=0xFF)
    pairs = flat.view(np.uint64).reshape(N, 2)  # [key, prefix]
    got = oracle.radix_sort_key_prefix(pairs.copy(), sb, eb, desc, signed)
    perm = expected_perm(pairs[:, 1].copy().view(np.int64), sb, eb, desc, signed)
    exp = pairs[perm]
    assert (got == exp).all()


def random_bitmask(rand: "oracle.XorShiftRandom") -> int:
    # RadixSortSuite.scala:138-145 randomBitMask
    tmp = -1
    for _ in range(rand.next_int(5) + 1):
        tmp &= rand.next_long()
    return tmp


@pytest.mark.parametrize("name,sb,eb,desc,signed", SORT_TYPES)
@pytest.mark.parametrize("seed", [1, 2, 3, 4, 5])
def test_fuzz_bitmask(name, sb, eb, desc, signed, seed):
    # RadixSortSuite.scala:178-188 fuzz, with fixed seeds instead of nanoTime
    rand = oracle.XorShiftRandom(seed)
    mask = random_bitmask(rand)
    data = rand.fill_longs(N, mask=mask)
    got = oracle.radix_sort_longs(data, sb, eb, desc, signed).view(np.int64)
    exp = data[expected_perm(data, sb, eb, desc, signed)]
    assert (got == exp).all()


@pytest.mark.parametrize("seed", [11, 12, 13])
def test_fuzz_key_prefix_bitmask(seed):
    rand = oracle.XorShiftRandom(seed)
    mask = random_bitmask(rand)
    flat = rand.fill_longs(2 * N, mask=mask)
    pairs = flat.view(np.uint64).reshape(N, 2)
    for name, sb, eb, desc, signed in SORT_TYPES:
        got = oracle.radix_sort_key_prefix(pairs.copy(), sb, eb, desc, signed)
        perm = expected_perm(pairs[:, 1].copy().view(np.int64), sb, eb, desc, signed)
        assert (got == pairs[perm]).all()


# --- operator-level sort oracle (SortExec semantics) ---

def py_prefix_double(v: float) -> int:
    """Independent Python restatement of DoublePrefixComparator.computePrefix
    (PrefixComparators.java:72-83) to cross-check the C one."""
    import struct
    if v == 0.0:
        v = 0.0  # normalizes -0.0
    if v != v:
        bits = 0x7FF8000000000000
    else:
        bits = struct.unpack("<Q", struct.pack("<d", v))[0]
    mask = (0xFFFFFFFFFFFFFFFF if bits >> 63 else 0) | 0x8000000000000000
    return (bits ^ mask) & 0xFFFFFFFFFFFFFFFF


def test_prefix_double_cross_impl():
    vals = [0.0, -0.0, 1.5, -1.5, float("inf"), float("-inf"), float("nan"),
            5e-324, -5e-324, 1e308, -1e308]
    for v in vals:
        assert oracle.prefix_double(v) == py_prefix_double(v)
    # monotonicity: prefix order == IEEE total order (with -0.0 == 0.0)
    xs = np.array([-np.inf, -1e308, -2.5, -1.0, -5e-324, -0.0, 0.0, 5e-324,
                   1.0, 2.5, 1e308, np.inf, np.nan])
    ps = np.array([oracle.prefix_double(float(x)) for x in xs], dtype=np.uint64)
    assert (np.diff(ps.astype(object)) >= 0).all()
    assert ps[5] == ps[6]  # -0.0 == 0.0


def test_sort_perm_i64_with_nulls():
    rng = np.random.default_rng(0)
    n = 5000
    keys = rng.integers(-100, 100, n).astype(np.int64)
    validity = np.packbits(rng.random(n) > 0.1, bitorder="little")
    valid = np.unpackbits(validity, count=n, bitorder="little").astype(bool)
    for desc in [False, True]:
        perm = oracle.sort_perm(keys, desc=desc, validity=validity)
        nulls_first = not desc  # SortOrder defaults
        null_rows = np.flatnonzero(~valid)
        valid_rows = np.flatnonzero(valid)
        vk = keys[valid_rows].view(np.uint64) ^ SIGN
        if desc:
            vk = ~vk
        sorted_valid = valid_rows[np.argsort(vk, kind="stable")]
        exp = np.concatenate([null_rows, sorted_valid] if nulls_first
                             else [sorted_valid, null_rows])
        assert (perm == exp).all()


def test_sort_perm_f64_special_values():
    keys = np.array([1.5, -0.0, np.nan, 0.0, -np.inf, np.inf, 2.5, np.nan, -1.0])
    perm = oracle.sort_perm(keys)
    got = keys[perm]
    # ascending: -inf, -1, (-0.0, 0.0 stable), 1.5, 2.5, inf, nan, nan
    assert got[0] == -np.inf and got[1] == -1.0
    assert got[2] == 0.0 and np.signbit(got[2])   # -0.0 kept before 0.0 (stable tie)
    assert got[3] == 0.0 and not np.signbit(got[3])
    assert got[4] == 1.5 and got[5] == 2.5 and got[6] == np.inf
    assert np.isnan(got[7]) and np.isnan(got[8])
    # descending: nans first (reverse binary order), then inf ...
    permd = oracle.sort_perm(keys, desc=True)
    gotd = keys[permd]
    assert np.isnan(gotd[0]) and np.isnan(gotd[1]) and gotd[2] == np.inf
    assert gotd[-1] == -np.inf


def test_xorshift_restatement_lock():
    """Regression lock for the XORShiftRandom restatement (fixture header
    explains provenance)."""
    import json, os
    fix = json.load(open(os.path.join(os.path.dirname(__file__), "golden",
                                      "xorshift_restatement_lock.json")))
    assert oracle.XorShiftRandom(123).fill_longs(64).tolist() == \
        fix["seed123_nextLong_first64"]
    assert oracle.XorShiftRandom(123).fill_longs(32, mask=0xFF).tolist() == \
        fix["seed123_masked_ff_first32"]
    r = oracle.XorShiftRandom(1)
    assert [r.next_int(5) for _ in range(16)] == fix["seed1_nextInt5_first16"]


def test_mt_sort_matches_single_thread():
    # the OpenMP baseline leg must produce the identical stable permutation
    keys = oracle.gen_i64(9, 500_000)
    assert (oracle.sort_perm_mt(keys, 4) == oracle.sort_perm(keys)).all()
    low = oracle.gen_i64(10, 100_000, range_=50)  # heavy ties: stability
    assert (oracle.sort_perm_mt(low, 8) == oracle.sort_perm(low)).all()
