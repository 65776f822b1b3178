"""Parquet scan feed: pyarrow -> device ColumnarBatch -> the real TPC-H Q1
plan through the rule (config-5 substance: decimal(12,2) as scaled int64,
date32 days, dictionary-encoded flag columns, composite (returnflag,
linestatus) grouping via the narrow-tuple pack rule)."""
import datetime
import decimal

import numpy as np
import pytest

pa = pytest.importorskip("pyarrow")
import pyarrow.parquet as pq  # noqa: E402

from spark_amd.parquet_io import arrow_column_to_numpy  # noqa: E402


def test_arrow_conversion_cpu(tmp_path):
    """host-side conversions (no GPU): ints, floats, date32, decimal(12,2)
    scaled int64, dictionary strings, NULL bitmaps."""
    ints = pa.array([1, None, -3, 2 ** 40], type=pa.int64())
    vals, bitmap, d = arrow_column_to_numpy(ints)
    assert d is None and vals.dtype == np.int64
    assert list(np.unpackbits(bitmap, count=4, bitorder="little")) == [1, 0, 1, 1]
    assert vals[0] == 1 and vals[2] == -3 and vals[3] == 2 ** 40

    dec = pa.array([decimal.Decimal("12345.67"), None,
                    decimal.Decimal("-0.03")], type=pa.decimal128(12, 2))
    vals, bitmap, _ = arrow_column_to_numpy(dec)
    assert vals[0] == 1234567 and vals[2] == -3  # scaled by 10^2

    dates = pa.array([datetime.date(1995, 3, 15), datetime.date(1970, 1, 2)],
                     type=pa.date32())
    vals, bitmap, _ = arrow_column_to_numpy(dates)
    assert vals[1] == 1 and bitmap is None
    assert vals[0] == (datetime.date(1995, 3, 15)
                       - datetime.date(1970, 1, 1)).days

    strs = pa.array(["N", "A", "N", "R"]).dictionary_encode()
    vals, bitmap, d = arrow_column_to_numpy(strs)
    assert [d[i] for i in vals] == ["N", "A", "N", "R"]

    f = pa.array([1.5, None], type=pa.float32())
    vals, bitmap, _ = arrow_column_to_numpy(f)
    assert vals.dtype == np.float64 and vals[0] == 1.5


def make_lineitem(path, n=200_000, seed=7):
    rng = np.random.default_rng(seed)
    qty = rng.integers(100, 50_00, n)            # decimal(12,2) scaled
    price = rng.integers(90_000, 10_500_000, n)
    disc = rng.integers(0, 11, n)                # 0.00 .. 0.10
    tax = rng.integers(0, 9, n)
    ship = rng.integers(8035, 10591, n)          # days: 1992..1998-12-25
    rf = rng.integers(0, 3, n)                   # A/N/R
    ls = rng.integers(0, 2, n)                   # F/O
    def dec_col(a):
        return pa.array([decimal.Decimal(int(v)) / 100 for v in a],
                        type=pa.decimal128(12, 2))
    tbl = pa.table({
        "l_quantity": dec_col(qty),
        "l_extendedprice": dec_col(price),
        "l_discount": dec_col(disc),
        "l_tax": dec_col(tax),
        "l_shipdate": pa.array(
            (np.asarray(ship, dtype="int64")).astype("datetime64[D]")),
        "l_returnflag": pa.array(np.array(["A", "N", "R"])[rf])
            .dictionary_encode(),
        "l_linestatus": pa.array(np.array(["F", "O"])[ls])
            .dictionary_encode(),
    })
    pq.write_table(tbl, path, row_group_size=n // 2 + 1)
    return dict(qty=qty, price=price, disc=disc, tax=tax, ship=ship,
                rf=rf, ls=ls)


@pytest.mark.gpu
def test_q1_over_parquet_scan(tmp_path):
    """the real Q1 aggregate set over a lineitem-shaped Parquet file,
    through the injected rule: filter (shipdate <= cutoff) -> projected
    decimal products (scale-4 disc_price, scale-6 charge as exact scaled
    int64) -> GROUP BY (returnflag, linestatus) with 4 SUMs + AVG +
    COUNT(*) -> ORDER BY the grouping keys."""
    import torch
    from spark_amd import exec as gx

    path = str(tmp_path / "lineitem.parquet")
    ref = make_lineitem(path)
    cutoff = 10_470   # ~1998-09-02 (date '1998-12-01' - 90 days)

    # the scan feeds the plan directly: one batch per row group streams
    # through the filter; the aggregate accumulates the whole partition
    scan_node = gx.GpuParquetScanExec(path)
    probe = list(gx.GpuParquetScanExec(path).execute_columnar())
    assert len(probe) == 2   # two row groups
    for b in probe:
        b.close()

    filt = gx.FilterExec("l_shipdate", "<=", cutoff, scan_node)
    proj = gx.ProjectExec(
        ["l_returnflag", "l_linestatus", "l_quantity", "l_extendedprice",
         "l_discount",
         ("__one_minus_disc", "l_discount", "rsub", None, 100),
         ("__one_plus_tax", "l_tax", "+", None, 100)], filt)
    proj2 = gx.ProjectExec(
        ["l_returnflag", "l_linestatus", "l_quantity", "l_extendedprice",
         "l_discount", "__one_plus_tax",
         ("disc_price", "l_extendedprice", "*", "__one_minus_disc", None)],
        proj)
    proj3 = gx.ProjectExec(
        ["l_returnflag", "l_linestatus", "l_quantity", "l_extendedprice",
         "l_discount", "disc_price",
         ("charge", "disc_price", "*", "__one_plus_tax", None)], proj2)
    agg = gx.HashAggregateExec(
        ("l_returnflag", "l_linestatus"),
        [("sum", "l_quantity"), ("sum", "l_extendedprice"),
         ("sum", "disc_price"), ("sum", "charge"),
         ("avg", "l_discount"), ("count*", None)],
        "complete", proj3)
    plan = gx.SortExec([gx.SortOrder("l_returnflag"),
                        gx.SortOrder("l_linestatus")], False, agg)
    gpu = gx.GpuColumnarRule().pre_columnar_transitions(plan)
    res = list(gpu.execute_columnar())
    assert len(res) == 1
    r = res[0]

    # independent numpy expectation from the generator arrays
    m = ref["ship"] <= cutoff
    qty, price, disc, tax = (ref[k][m] for k in ("qty", "price", "disc", "tax"))
    rf, ls = ref["rf"][m], ref["ls"][m]
    disc_price = price * (100 - disc)            # scale 4
    charge = disc_price * (100 + tax)            # scale 6
    dicts = scan_node.dictionaries
    got_rf = r.column("l_returnflag").cpu().numpy()
    got_ls = r.column("l_linestatus").cpu().numpy()
    assert len(got_rf) == len(set(zip(rf.tolist(), ls.tolist())))
    for gi in range(len(got_rf)):
        # grouping keys decode through the parquet dictionaries (ids are
        # in the file dictionary's first-occurrence order)
        a = ["A", "N", "R"].index(dicts["l_returnflag"][got_rf[gi]])
        b = ["F", "O"].index(dicts["l_linestatus"][got_ls[gi]])
        sel = (rf == a) & (ls == b)
        assert r.column("sum(l_quantity)").cpu().numpy()[gi] == qty[sel].sum()
        assert (r.column("sum(l_extendedprice)").cpu().numpy()[gi]
                == price[sel].sum())
        assert (r.column("sum(disc_price)").cpu().numpy()[gi]
                == disc_price[sel].sum())
        assert r.column("sum(charge)").cpu().numpy()[gi] == charge[sel].sum()
        np.testing.assert_allclose(
            r.column("avg(l_discount)").cpu().numpy()[gi],
            disc[sel].mean(), rtol=1e-9)
        assert r.column("count(1)").cpu().numpy()[gi] == sel.sum()
    # ordered by the grouping-key ids (what the plan's SortExec was given;
    # a host ordering by the STRINGS would sort or remap the dictionary)
    assert (np.diff(got_rf * 2 + got_ls) > 0).all()
