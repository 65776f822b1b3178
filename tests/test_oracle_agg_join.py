"""Check the oracle's GROUP BY / inner-join restatements against independent
numpy/dict computations (semantics cites in oracle/oracle.c; parity for these
two operators is pinned by construction — SURVEY.md §8(c))."""
import numpy as np

import oracle


def unpack(validity, n):
    return np.unpackbits(validity, count=n, bitorder="little").astype(bool)


def test_hash_agg_matches_numpy():
    rng = np.random.default_rng(1)
    n = 100_000
    keys = rng.integers(0, 1000, n).astype(np.int64)
    vals = rng.random(n)
    ok, okv, osum, osv, ocnt = oracle.hash_agg(keys, vals)
    assert okv.all() and osv.all()
    # first-occurrence order
    _, first_idx = np.unique(keys, return_index=True)
    exp_keys = keys[np.sort(first_idx)]
    assert (ok == exp_keys).all()
    # sums/counts per key (order-insensitive compare, 1e-6 relative on SUM —
    # the north-star tolerance; oracle accumulation order == input order)
    order = np.argsort(ok, kind="stable")
    uk = np.unique(keys)
    exp_sum = np.zeros(len(uk))
    exp_cnt = np.zeros(len(uk), dtype=np.int64)
    inv = np.searchsorted(uk, keys)
    np.add.at(exp_sum, inv, vals)
    np.add.at(exp_cnt, inv, 1)
    assert (ok[order] == uk).all()
    assert (ocnt[order] == exp_cnt).all()
    np.testing.assert_allclose(osum[order], exp_sum, rtol=1e-9)


def test_hash_agg_nulls():
    keys = np.array([1, 2, 1, 3, 2, 1], dtype=np.int64)
    vals = np.array([1.0, 2.0, 3.0, 4.0, 5.0, 6.0])
    key_validity = np.packbits([1, 1, 0, 1, 1, 1], bitorder="little")  # row2 null key
    val_validity = np.packbits([1, 0, 1, 1, 1, 1], bitorder="little")  # row1 null val
    ok, okv, osum, osv, ocnt = oracle.hash_agg(keys, vals, key_validity, val_validity)
    # groups in first-occurrence order: 1, 2, NULL, 3
    assert len(ok) == 4
    assert ok[0] == 1 and okv[0] == 1
    assert ok[1] == 2 and okv[1] == 1
    assert okv[2] == 0               # NULL-key group (NULL is a valid group key)
    assert ok[3] == 3 and okv[3] == 1
    assert osum[0] == 7.0 and ocnt[0] == 2       # rows 0,5 (row2 went to NULL group)
    assert osv[1] == 1 and osum[1] == 5.0 and ocnt[1] == 1  # row1 val NULL skipped
    assert osum[2] == 3.0 and ocnt[2] == 1       # null-key group got row2's value
    assert osum[3] == 4.0 and ocnt[3] == 1


def test_hash_agg_all_null_vals_gives_null_sum():
    keys = np.array([7, 7], dtype=np.int64)
    vals = np.array([1.0, 2.0])
    val_validity = np.packbits([0, 0], bitorder="little")
    ok, okv, osum, osv, ocnt = oracle.hash_agg(keys, vals, None, val_validity)
    # SUM of all-NULL inputs is NULL (Sum.scala:134-141), COUNT is 0
    assert len(ok) == 1 and osv[0] == 0 and ocnt[0] == 0


def test_join_inner_matches_dict():
    rng = np.random.default_rng(2)
    bn, pn = 2000, 3000
    bkeys = rng.integers(0, 500, bn).astype(np.int64)
    pkeys = rng.integers(0, 700, pn).astype(np.int64)
    op, ob = oracle.join_inner(bkeys, pkeys)
    exp = []
    from collections import defaultdict
    idx = defaultdict(list)
    for i, k in enumerate(bkeys.tolist()):
        idx[k].append(i)
    for i, k in enumerate(pkeys.tolist()):
        for b in idx.get(k, ()):
            exp.append((i, b))
    got = sorted(zip(op.tolist(), ob.tolist()))
    assert got == sorted(exp)
    # oracle emit order: probe order, then build insertion order
    assert got == list(zip(op.tolist(), ob.tolist()))


def test_join_null_keys_never_match():
    bkeys = np.array([1, 2], dtype=np.int64)
    pkeys = np.array([1, 2], dtype=np.int64)
    bvalid = np.packbits([1, 0], bitorder="little")
    pvalid = np.packbits([0, 1], bitorder="little")
    op, ob = oracle.join_inner(bkeys, pkeys, bvalid, pvalid)
    assert len(op) == 0  # 1 matches only null-probe row, 2 only null-build row


def test_join_duplicates_both_sides():
    bkeys = np.array([5, 5, 5], dtype=np.int64)
    pkeys = np.array([5, 5], dtype=np.int64)
    op, ob = oracle.join_inner(bkeys, pkeys)
    assert len(op) == 6
    assert sorted(zip(op.tolist(), ob.tolist())) == [
        (0, 0), (0, 1), (0, 2), (1, 0), (1, 1), (1, 2)]


def test_baseline_config1_groupby_sum_1m():
    """BASELINE configs[0]: local[2] df.groupBy(key).agg(sum(val)) on a
    1M-row 2-col DataFrame — the CPU-runnable case, checked through the
    oracle restatement with the checkAnswer-style order-insensitive compare."""
    n = 1_000_000
    keys = oracle.gen_i64(1042, n, range_=10_000)
    vals = oracle.gen_f64_unit(1043, n)
    ok, okv, osum, osv, ocnt = oracle.hash_agg(keys, vals)
    assert okv.all() and osv.all()
    # independent check: numpy groupby
    uk = np.unique(keys)
    exp = np.zeros(len(uk))
    np.add.at(exp, np.searchsorted(uk, keys), vals)
    order = np.argsort(ok)
    assert (ok[order] == uk).all()
    np.testing.assert_allclose(osum[order], exp, rtol=1e-9)
    assert int(ocnt.sum()) == n
