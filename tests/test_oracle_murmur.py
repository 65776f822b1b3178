"""Pin the oracle's Murmur3/partition-id restatement to the reference's own
known-answer vectors (Murmur3_x86_32Suite.java:39-56) and Pmod semantics
(partitioning.scala:328-330)."""
import json
import os

import numpy as np

import oracle

GOLDEN = os.path.join(os.path.dirname(__file__), "golden", "murmur3_known_answers.json")


def test_known_integer_inputs():
    vec = json.load(open(GOLDEN))
    for x, expected in vec["hashInt"]:
        assert oracle.hash_int(x, 0) == expected


def test_known_long_inputs():
    vec = json.load(open(GOLDEN))
    for x, expected in vec["hashLong"]:
        assert oracle.hash_long(x, 0) == expected


def test_hash_bytes2_matches_hash_via_words():
    # hashUnsafeBytes2 with length 8 over the LE bytes of a long equals
    # hashLong (both process two LE 32-bit words) — internal consistency
    # mirroring Murmur3_x86_32.java:109-122 vs :84-95.
    for v in [0, 1, -1, 42, -42, 2**63 - 1, -(2**63)]:
        b = int(v).to_bytes(8, "little", signed=True)
        assert oracle.hash_bytes2(b, 7) == oracle.hash_long(v, 7)


def test_pmod():
    # Pmod(a, n) = ((a % n) + n) % n (catalyst arithmetic.scala Pmod)
    for a in [-7, -1, 0, 1, 7, -(2**31), 2**31 - 1]:
        for n in [1, 2, 8, 200]:
            assert oracle.pmod(a, n) == ((a % n) + n) % n


def test_partition_ids_match_definition():
    # partitionIdExpression = Pmod(Murmur3Hash(keys, 42), n)
    # (partitioning.scala:328-330)
    keys = np.array([0, 1, -1, 42, 123456789, -(2**63), 2**63 - 1], dtype=np.int64)
    pids = oracle.partition_ids(keys, 8)
    for k, p in zip(keys.tolist(), pids.tolist()):
        assert p == oracle.pmod(oracle.hash_long(k, 42), 8)


def test_partition_ids_null_passes_seed_through():
    # NULL key leaves the running hash at the seed (hash.scala HashExpression.eval)
    keys = np.array([5, 7], dtype=np.int64)
    validity = np.array([0b10], dtype=np.uint8)  # row0 null, row1 valid
    pids = oracle.partition_ids(keys, 8, validity=validity)
    assert pids[0] == oracle.pmod(42, 8)
    assert pids[1] == oracle.pmod(oracle.hash_long(7, 42), 8)


def test_partition_ids_multi_column_chaining():
    # Murmur3Hash(c1, c2, seed 42) chains: h = hashLong(c2, hashLong(c1, 42))
    # (hash.scala:849-860); NULL columns leave the running hash unchanged.
    c1 = np.array([1, 2, 3], dtype=np.int64)
    c2 = np.array([10, 20, 30], dtype=np.int64)
    pids = oracle.partition_ids_multi([c1, c2], 8)
    for i in range(3):
        h = oracle.hash_long(int(c2[i]), oracle.hash_long(int(c1[i]), 42))
        assert pids[i] == oracle.pmod(h, 8)
    # single column must agree with the single-column path
    assert (oracle.partition_ids_multi([c1], 8) == oracle.partition_ids(c1, 8)).all()
    # a NULL in column 1 skips that link in the chain
    n = 3
    vb = np.zeros((2, (n + 7) // 8), dtype=np.uint8)
    vb[0] = np.packbits([0, 1, 1], bitorder="little")   # c1 row0 NULL
    vb[1] = np.packbits([1, 1, 1], bitorder="little")
    pids = oracle.partition_ids_multi([c1, c2], 8, validity=np.ascontiguousarray(vb))
    assert pids[0] == oracle.pmod(oracle.hash_long(10, 42), 8)
