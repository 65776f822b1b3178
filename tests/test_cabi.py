"""CPU-side checks of the C-ABI boundary: libgpuq.so builds for gfx950, loads
in a GPU-less container, and exports every symbol include/gpuq.h declares
(no compute calls without a GPU)."""
import ctypes
import os
import re
import subprocess

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SO = os.path.join(ROOT, "spark_amd", "libgpuq.so")
HDR = os.path.join(ROOT, "include", "gpuq.h")


def _ensure_built():
    if not os.path.exists(SO):
        subprocess.run(["make", "-C", os.path.join(ROOT, "spark_amd", "csrc"), "-s"],
                       check=True)


def header_symbols():
    syms = []
    for line in open(HDR):
        m = re.match(r"^(?:const char\*|int|int64_t)\s+(gpuq_\w+)\(", line)
        if m:
            syms.append(m.group(1))
    return syms


def test_header_declares_expected_entry_points():
    syms = header_symbols()
    for s in ["gpuq_last_error", "gpuq_sort_perm", "gpuq_hash_agg_i64_f64",
              "gpuq_partition_perm", "gpuq_join_build_i64", "gpuq_join_probe_i64",
              "gpuq_gather"]:
        assert s in syms


def test_so_loads_and_exports_all_header_symbols():
    _ensure_built()
    L = ctypes.CDLL(SO)
    for s in header_symbols():
        assert hasattr(L, s), f"libgpuq.so missing export {s}"
    # error string callable without a GPU
    L.gpuq_last_error.restype = ctypes.c_char_p
    assert isinstance(L.gpuq_last_error(), bytes)


def test_so_is_gfx950_only():
    _ensure_built()
    out = subprocess.run(
        ["/opt/rocm/lib/llvm/bin/llvm-objdump", "--offloading", SO],
        capture_output=True, text=True)
    blob = out.stdout + out.stderr
    if "gfx" in blob:
        assert "gfx950" in blob
        assert not re.search(r"gfx(?!950)\d+", blob), "non-gfx950 code objects found"


def test_python_binding_raises_without_gpu():
    _ensure_built()
    import torch
    if torch.cuda.is_available():
        pytest.skip("GPU present")
    from spark_amd import gpuq
    with pytest.raises(Exception):
        gpuq.gen_i64(1, 10)  # no CPU fallback: must raise, not compute


def test_argument_validation_without_gpu():
    """The C-ABI validates arguments before any HIP call, so contract
    violations fail loudly even in a GPU-less container."""
    import ctypes
    _ensure_built()
    L = ctypes.CDLL(SO)
    L.gpuq_last_error.restype = ctypes.c_char_p

    class Col(ctypes.Structure):
        _fields_ = [("data", ctypes.c_void_p), ("validity", ctypes.c_void_p),
                    ("dtype", ctypes.c_int32)]

    col_i64 = Col(None, None, 0)
    # sort: rows beyond u32 rowids
    L.gpuq_sort_perm.restype = ctypes.c_int32
    L.gpuq_sort_perm.argtypes = [ctypes.c_void_p, ctypes.c_int64, Col,
                                 ctypes.c_int32, ctypes.c_int32, ctypes.c_void_p,
                                 ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int64]
    rc = L.gpuq_sort_perm(None, 1 << 33, col_i64, 0, 1, None, None, None, 0)
    assert rc == 2 and b"2^32" in L.gpuq_last_error()
    # partition: too many partitions
    L.gpuq_partition_perm.restype = ctypes.c_int32
    L.gpuq_partition_perm.argtypes = [ctypes.c_void_p, ctypes.c_int64, Col,
                                      ctypes.c_int32, ctypes.c_void_p,
                                      ctypes.c_void_p, ctypes.c_void_p,
                                      ctypes.c_int64]
    rc = L.gpuq_partition_perm(None, 10, col_i64, 1 << 17, None, None, None, 0)
    assert rc == 2 and b"65536" in L.gpuq_last_error()
    # agg: capacity not a power of two
    L.gpuq_hash_agg_i64_f64.restype = ctypes.c_int32
    rc = L.gpuq_hash_agg_i64_f64(None, ctypes.c_int64(10), col_i64, col_i64,
                                 None, ctypes.c_int64(1000),
                                 1, 1, 3, None, None, None, None, None, None)
    assert rc == 2 and b"power of two" in L.gpuq_last_error()
    # join: build side beyond 31-bit rowids
    L.gpuq_join_build_i64.restype = ctypes.c_int32
    L.gpuq_join_build_i64.argtypes = [ctypes.c_void_p, ctypes.c_int64, Col,
                                      ctypes.c_void_p, ctypes.c_int64]
    rc = L.gpuq_join_build_i64(None, (1 << 31), col_i64, None, 1 << 32)
    assert rc == 2 and b"31-bit" in L.gpuq_last_error()
