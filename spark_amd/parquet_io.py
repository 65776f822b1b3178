"""Parquet -> device ColumnarBatch ingestion (the scan feed, SURVEY
§8(f).1 full form).

The reference's FileSourceScanExec already emits ColumnarBatch when every
column is vectorizable (DataSourceScanExec.scala:735 supportsColumnar;
VectorizedParquetRecordReader.java:67 reads row groups into
ColumnVectors). This module is the host half a JNI deployment would run:
pyarrow decodes a row group, columns are normalized to the engine's types
and shipped to HBM with their Arrow validity bitmaps:

- int8/16/32/64            -> int64
- float32/64               -> float64
- date32                   -> int64 days since epoch (DateType's physical
                              int, DateTimeUtils.scala)
- decimal128(p<=18, s)     -> int64 scaled by 10^s (Spark's compact-long
                              Decimal representation, Decimal.scala:
                              "compact" longs for precision <= 18)
- dictionary<string>       -> int64 dictionary ids + the dictionary
                              (engine operates on ids; ids are assigned in
                              the parquet dictionary's order)

NULLs travel as Arrow validity bitmaps (LSB-first), bit-identical to the
engine's layout.
"""
from typing import Dict, List, Optional, Tuple

import numpy as np


def _decimal_to_scaled_i64(arr) -> np.ndarray:
    """decimal128 values are 16-byte little-endian scaled integers; for
    precision <= 18 the value fits the low 8 bytes (high word is sign
    extension) — exactly Spark's compact-long Decimal."""
    import pyarrow as pa
    buf = arr.buffers()[1]
    off = arr.offset
    raw = np.frombuffer(buf, dtype=np.int64,
                        count=2 * (off + len(arr)))[2 * off:]
    pairs = raw.reshape(-1, 2)
    lo, hi = pairs[:, 0].copy(), pairs[:, 1]
    if not ((hi == (lo >> 63)) | np.asarray(arr.is_null())).all():
        raise ValueError("decimal128 value exceeds 18 digits (compact long)")
    return lo


def arrow_column_to_numpy(arr) -> Tuple[np.ndarray, Optional[np.ndarray],
                                        Optional[List[str]]]:
    """-> (values int64/float64, validity bitmap bytes or None,
    dictionary or None). Data under NULLs is unspecified (validity
    governs), matching the ColumnVector contract."""
    import pyarrow as pa
    if arr.null_count:
        # arrow validity bitmap is LSB-first like ours; account for offset
        # by round-tripping through numpy bools
        mask = ~np.asarray(arr.is_null())
        bitmap = np.packbits(mask, bitorder="little")
    else:
        bitmap = None
    t = arr.type
    dictionary = None
    if pa.types.is_dictionary(t):
        dictionary = arr.dictionary.to_pylist()
        vals = np.asarray(arr.indices.fill_null(0)).astype(np.int64)
    elif pa.types.is_decimal(t):
        if t.precision > 18:
            raise ValueError(f"decimal precision {t.precision} > 18")
        vals = _decimal_to_scaled_i64(arr)
    elif pa.types.is_date32(t):
        vals = np.asarray(arr.fill_null(0)).astype("datetime64[D]") \
            .astype(np.int64)
    elif pa.types.is_floating(t):
        vals = np.asarray(arr.fill_null(0)).astype(np.float64)
    elif pa.types.is_integer(t):
        vals = np.asarray(arr.fill_null(0)).astype(np.int64)
    else:
        raise ValueError(f"unsupported parquet type {t}")
    return np.ascontiguousarray(vals), bitmap, dictionary


def read_row_group_to_device(pf, rg: int, columns=None):
    """Read one row group and ship it to the current CUDA device.
    Returns (cols: name->tensor, validity: name->bitmap tensor,
    dictionaries: name->list)."""
    import torch
    tbl = pf.read_row_group(rg, columns=columns)
    cols, validity, dicts = {}, {}, {}
    for name in tbl.column_names:
        arr = tbl.column(name).combine_chunks()
        vals, bitmap, dictionary = arrow_column_to_numpy(arr)
        host = torch.from_numpy(vals)
        if torch.cuda.is_available():
            host = host.pin_memory()
            dev = host.cuda(non_blocking=True)
        else:
            dev = host
        cols[name] = dev
        if bitmap is not None:
            bt = torch.from_numpy(bitmap)
            validity[name] = bt.cuda(non_blocking=True) \
                if torch.cuda.is_available() else bt
        if dictionary is not None:
            dicts[name] = dictionary
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    return cols, validity, dicts
