"""CPU oracle for the GPU engine — TEST INFRASTRUCTURE ONLY.

ctypes wrapper over oracle/liboracle.so (built by `make -C oracle`, also by
__graft_entry__.build()). Only tests/, __graft_entry__.smoke() and bench.py's
cpu_baseline leg may import this package, and only as the parity checker /
reported CPU baseline — never as a shipped compute path. The product path
(spark_amd) fails loudly when its HIP extension is missing; it never falls
back here.

Semantics restated (with file:line cites) in oracle/oracle.c.
"""
import ctypes
import os
import subprocess

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "liboracle.so")


def build(force: bool = False) -> None:
    if force or not os.path.exists(_SO) or (
        os.path.getmtime(_SO) < os.path.getmtime(os.path.join(_DIR, "oracle.c"))
    ):
        subprocess.run(["make", "-C", _DIR, "-s"], check=True)


_lib = None


def lib() -> ctypes.CDLL:
    global _lib
    if _lib is None:
        build()
        _lib = ctypes.CDLL(_SO)
        L = _lib
        i32, i64, u64, f64, b = (
            ctypes.c_int32, ctypes.c_int64, ctypes.c_uint64,
            ctypes.c_double, ctypes.c_bool)
        P = ctypes.POINTER
        L.mm3_hash_int.restype = i32
        L.mm3_hash_int.argtypes = [i32, i32]
        L.mm3_hash_long.restype = i32
        L.mm3_hash_long.argtypes = [i64, i32]
        L.mm3_hash_bytes2.restype = i32
        L.mm3_hash_bytes2.argtypes = [ctypes.c_void_p, i32, i32]
        L.spark_pmod.restype = i32
        L.spark_pmod.argtypes = [i32, i32]
        L.prefix_double.restype = u64
        L.prefix_double.argtypes = [f64]
        L.oracle_partition_ids_i64.restype = None
        L.oracle_partition_ids_i64.argtypes = [ctypes.c_void_p, ctypes.c_void_p, i64, i32, ctypes.c_void_p]
        L.oracle_partition_ids_i64_multi.restype = None
        L.oracle_partition_ids_i64_multi.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                                                     i32, i64, i32, ctypes.c_void_p]
        L.oracle_radix_sort_longs.restype = i64
        L.oracle_radix_sort_longs.argtypes = [ctypes.c_void_p, i64, i32, i32, b, b]
        L.oracle_radix_sort_key_prefix.restype = i64
        L.oracle_radix_sort_key_prefix.argtypes = [ctypes.c_void_p, i64, i32, i32, b, b]
        L.oracle_sort_perm_i64.restype = None
        L.oracle_sort_perm_i64.argtypes = [ctypes.c_void_p, ctypes.c_void_p, i64, b, b, ctypes.c_void_p]
        L.oracle_sort_perm_f64.restype = None
        L.oracle_sort_perm_f64.argtypes = [ctypes.c_void_p, ctypes.c_void_p, i64, b, b, ctypes.c_void_p]
        L.oracle_hash_agg_i64_f64.restype = i64
        L.oracle_hash_agg_i64_f64.argtypes = [ctypes.c_void_p] * 4 + [i64] + [ctypes.c_void_p] * 5
        L.oracle_join_inner_i64.restype = i64
        L.oracle_join_inner_i64.argtypes = [ctypes.c_void_p, ctypes.c_void_p, i64,
                                            ctypes.c_void_p, ctypes.c_void_p, i64,
                                            ctypes.c_void_p, ctypes.c_void_p, i64]
        L.oracle_sort_perm_i64_mt.restype = None
        L.oracle_sort_perm_i64_mt.argtypes = [ctypes.c_void_p, i64, ctypes.c_void_p, i32]
        L.oracle_gather_i64_mt.restype = None
        L.oracle_gather_i64_mt.argtypes = [ctypes.c_void_p, ctypes.c_void_p, i64, ctypes.c_void_p]
        L.xorshift_state_init.restype = i64
        L.xorshift_state_init.argtypes = [i64]
        L.xorshift_state_next_long.restype = i64
        L.xorshift_state_next_long.argtypes = [P(i64)]
        L.xorshift_state_next_int.restype = i32
        L.xorshift_state_next_int.argtypes = [P(i64), i32]
        L.xorshift_fill_longs.restype = None
        L.xorshift_fill_longs.argtypes = [P(i64), ctypes.c_void_p, i64, i64]
        L.gen_u64.restype = u64
        L.gen_u64.argtypes = [u64, u64]
        L.gen_fill_u64.restype = None
        L.gen_fill_u64.argtypes = [u64, u64, i64, ctypes.c_void_p]
        L.gen_fill_i64_range.restype = None
        L.gen_fill_i64_range.argtypes = [u64, u64, i64, u64, ctypes.c_void_p]
        L.gen_fill_f64_unit.restype = None
        L.gen_fill_f64_unit.argtypes = [u64, u64, i64, ctypes.c_void_p]
    return _lib


def _ptr(a):
    return None if a is None else a.ctypes.data_as(ctypes.c_void_p)


def hash_int(x: int, seed: int = 42) -> int:
    return lib().mm3_hash_int(x, seed)


def hash_long(x: int, seed: int = 42) -> int:
    return lib().mm3_hash_long(x, seed)


def hash_bytes2(data: bytes, seed: int = 42) -> int:
    buf = np.frombuffer(data, dtype=np.uint8).copy()
    return lib().mm3_hash_bytes2(_ptr(buf), len(data), seed)


def pmod(a: int, n: int) -> int:
    return lib().spark_pmod(a, n)


def prefix_double(v: float) -> int:
    return lib().prefix_double(v)


def partition_ids(keys: np.ndarray, num_parts: int, validity=None) -> np.ndarray:
    keys = np.ascontiguousarray(keys, dtype=np.int64)
    out = np.empty(len(keys), dtype=np.int32)
    lib().oracle_partition_ids_i64(_ptr(keys), _ptr(validity), len(keys), num_parts, _ptr(out))
    return out


def partition_ids_multi(key_cols, num_parts: int, validity=None) -> np.ndarray:
    """Partition ids for multi-column keys (hash chains column-wise,
    hash.scala:849-860). key_cols: list of int64 arrays; validity: optional
    stacked bitmaps [ncols][ceil(n/8)]."""
    cols = np.ascontiguousarray(np.stack([np.asarray(c, dtype=np.int64)
                                          for c in key_cols]))
    n = cols.shape[1]
    out = np.empty(n, dtype=np.int32)
    lib().oracle_partition_ids_i64_multi(_ptr(cols), _ptr(validity), cols.shape[0],
                                         n, num_parts, _ptr(out))
    return out


def radix_sort_longs(vals: np.ndarray, start_byte=0, end_byte=7, desc=False, signed=False) -> np.ndarray:
    """Faithful RadixSort.sort (RadixSort.java:43) over int64 values."""
    n = len(vals)
    buf = np.zeros(2 * n, dtype=np.uint64)
    buf[:n] = vals.view(np.uint64)
    off = lib().oracle_radix_sort_longs(_ptr(buf), n, start_byte, end_byte, desc, signed)
    return buf[off:off + n].copy()


def radix_sort_key_prefix(pairs: np.ndarray, start_byte=0, end_byte=7, desc=False, signed=False) -> np.ndarray:
    """Faithful RadixSort.sortKeyPrefixArray (RadixSort.java:178).
    pairs: (n,2) uint64 [key, prefix]; sorted on prefix. Returns (n,2)."""
    n = pairs.shape[0]
    buf = np.zeros((2 * n, 2), dtype=np.uint64)
    buf[:n] = pairs
    off = lib().oracle_radix_sort_key_prefix(_ptr(buf), n, start_byte, end_byte, desc, signed)
    assert off % 2 == 0
    return buf[off // 2:off // 2 + n].copy()


def sort_perm(keys: np.ndarray, desc=False, nulls_first=None, validity=None) -> np.ndarray:
    """ORDER BY permutation for one int64/float64 column (SortExec semantics)."""
    if nulls_first is None:
        nulls_first = not desc  # SortOrder defaults (SortOrder.scala:35-45)
    out = np.empty(len(keys), dtype=np.int64)
    if keys.dtype == np.int64:
        lib().oracle_sort_perm_i64(_ptr(keys), _ptr(validity), len(keys), desc, nulls_first, _ptr(out))
    elif keys.dtype == np.float64:
        lib().oracle_sort_perm_f64(_ptr(keys), _ptr(validity), len(keys), desc, nulls_first, _ptr(out))
    else:
        raise TypeError(keys.dtype)
    return out


def hash_agg(keys: np.ndarray, vals: np.ndarray, key_validity=None, val_validity=None):
    """GROUP BY key -> (keys, key_valid, sums, sum_valid, counts) in
    first-occurrence order. COUNT here is COUNT(val) (non-null inputs)."""
    n = len(keys)
    keys = np.ascontiguousarray(keys, dtype=np.int64)
    vals = np.ascontiguousarray(vals, dtype=np.float64)
    ok = np.empty(n, dtype=np.int64)
    okv = np.empty(n, dtype=np.uint8)
    osum = np.empty(n, dtype=np.float64)
    osv = np.empty(n, dtype=np.uint8)
    ocnt = np.empty(n, dtype=np.int64)
    g = lib().oracle_hash_agg_i64_f64(_ptr(keys), _ptr(key_validity), _ptr(vals), _ptr(val_validity),
                                      n, _ptr(ok), _ptr(okv), _ptr(osum), _ptr(osv), _ptr(ocnt))
    return ok[:g].copy(), okv[:g].copy(), osum[:g].copy(), osv[:g].copy(), ocnt[:g].copy()


def join_inner(build_keys: np.ndarray, probe_keys: np.ndarray,
               build_validity=None, probe_validity=None):
    """Inner equi-join: returns (probe_rid, build_rid) int64 arrays."""
    bk = np.ascontiguousarray(build_keys, dtype=np.int64)
    pk = np.ascontiguousarray(probe_keys, dtype=np.int64)
    cnt = lib().oracle_join_inner_i64(_ptr(bk), _ptr(build_validity), len(bk),
                                      _ptr(pk), _ptr(probe_validity), len(pk),
                                      None, None, 0)
    op = np.empty(cnt, dtype=np.int64)
    ob = np.empty(cnt, dtype=np.int64)
    cnt2 = lib().oracle_join_inner_i64(_ptr(bk), _ptr(build_validity), len(bk),
                                       _ptr(pk), _ptr(probe_validity), len(pk),
                                       _ptr(op), _ptr(ob), cnt)
    assert cnt2 == cnt
    return op, ob


class XorShiftRandom:
    """Restatement of Spark's XORShiftRandom (XORShiftRandom.scala:36-68)
    for regenerating the reference's RadixSortSuite test data."""

    def __init__(self, seed: int):
        self._state = ctypes.c_int64(lib().xorshift_state_init(seed))

    def next_long(self) -> int:
        return lib().xorshift_state_next_long(ctypes.byref(self._state))

    def next_int(self, bound: int) -> int:
        return lib().xorshift_state_next_int(ctypes.byref(self._state), bound)

    def fill_longs(self, n: int, mask: int = -1) -> np.ndarray:
        out = np.empty(n, dtype=np.int64)
        lib().xorshift_fill_longs(ctypes.byref(self._state), _ptr(out), n, mask)
        return out


def gen_i64(seed: int, n: int, range_: int = 0, start: int = 0) -> np.ndarray:
    out = np.empty(n, dtype=np.int64)
    lib().gen_fill_i64_range(seed, start, n, range_, _ptr(out))
    return out


def gen_f64_unit(seed: int, n: int, start: int = 0) -> np.ndarray:
    out = np.empty(n, dtype=np.float64)
    lib().gen_fill_f64_unit(seed, start, n, _ptr(out))
    return out


def sort_perm_mt(keys: np.ndarray, nthreads: int = 0) -> np.ndarray:
    """OpenMP-parallel sort permutation (the multithreaded CPU-baseline leg;
    int64 asc non-null — same semantics as sort_perm)."""
    keys = np.ascontiguousarray(keys, dtype=np.int64)
    out = np.empty(len(keys), dtype=np.int64)
    lib().oracle_sort_perm_i64_mt(_ptr(keys), len(keys), _ptr(out), nthreads)
    return out


def gather_i64_mt(arr: np.ndarray, perm: np.ndarray) -> np.ndarray:
    out = np.empty(len(perm), dtype=np.int64)
    lib().oracle_gather_i64_mt(_ptr(np.ascontiguousarray(arr, dtype=np.int64)),
                               _ptr(np.ascontiguousarray(perm, dtype=np.int64)),
                               len(perm), _ptr(out))
    return out
