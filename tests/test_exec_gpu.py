"""GPU tests of the host operator mirror end-to-end: CPU-placeholder plans
rewritten by GpuColumnarRule, executed on device, parity-checked against the
oracle (single partition; the multi-rank exchange path is covered by the
gloo tests and the bench's RCCL leg)."""
import numpy as np
import pytest

import oracle

torch = pytest.importorskip("torch")
pytestmark = pytest.mark.gpu

from spark_amd import exec as gx  # noqa: E402


def dev_batch(**cols):
    return gx.ColumnarBatch({k: torch.from_numpy(v).cuda() for k, v in cols.items()})


def run_plan(plan):
    plan = gx.GpuColumnarRule().pre_columnar_transitions(plan)
    out = list(plan.execute_columnar())
    assert len(out) == 1
    return {k: v.cpu().numpy() for k, v in out[0].columns().items()}


def test_sort_plan():
    n = 100_000
    keys = oracle.gen_i64(seed=1, n=n, range_=500)
    pay = oracle.gen_f64_unit(seed=2, n=n)
    scan = gx.InputBatches([dev_batch(k=keys, v=pay)])
    got = run_plan(gx.SortExec(gx.SortOrder("k"), True, scan))
    perm = oracle.sort_perm(keys)
    assert (got["k"] == keys[perm]).all()
    assert (got["v"] == pay[perm]).all()


def test_agg_plan():
    n = 200_000
    keys = oracle.gen_i64(seed=3, n=n, range_=777)
    vals = oracle.gen_f64_unit(seed=4, n=n)
    scan = gx.InputBatches([dev_batch(k=keys, v=vals)])
    got = run_plan(gx.HashAggregateExec("k", [("sum", "v"), ("count", "v")],
                                        "complete", scan))
    ok, _, osum, _, ocnt = oracle.hash_agg(keys, vals)
    g, o = np.argsort(got["k"]), np.argsort(ok)
    assert (got["k"][g] == ok[o]).all()
    assert (got["count(v)"][g] == ocnt[o]).all()
    np.testing.assert_allclose(got["sum(v)"][g], osum[o], rtol=1e-6)


def test_join_plan():
    bn, pn = 50_000, 80_000
    bkeys = oracle.gen_i64(seed=5, n=bn, range_=40_000)
    bpay = oracle.gen_i64(seed=6, n=bn)
    pkeys = oracle.gen_i64(seed=7, n=pn, range_=40_000)
    ppay = oracle.gen_i64(seed=8, n=pn)
    left = gx.InputBatches([dev_batch(lk=pkeys, lp=ppay)])    # probe (stream)
    right = gx.InputBatches([dev_batch(rk=bkeys, rp=bpay)])   # build
    got = run_plan(gx.ShuffledHashJoinExec("lk", "rk", "right", left, right))
    op, ob = oracle.join_inner(bkeys, pkeys)
    assert len(got["lk"]) == len(op)
    g = np.lexsort((got["rp"], got["rk"], got["lp"], got["lk"]))
    exp_lk, exp_lp = pkeys[op], ppay[op]
    exp_rk, exp_rp = bkeys[ob], bpay[ob]
    o = np.lexsort((exp_rp, exp_rk, exp_lp, exp_lk))
    assert (got["lk"][g] == exp_lk[o]).all() and (got["lp"][g] == exp_lp[o]).all()
    assert (got["rk"][g] == exp_rk[o]).all() and (got["rp"][g] == exp_rp[o]).all()


def test_full_pipeline_join_then_agg_then_sort():
    """join -> aggregate on join output -> sort by key: exercises batch
    hand-off between GPU exec nodes staying on-device throughout."""
    bn, pn = 30_000, 60_000
    bkeys = oracle.gen_i64(seed=9, n=bn, range_=10_000)
    pkeys = oracle.gen_i64(seed=10, n=pn, range_=10_000)
    vals = oracle.gen_f64_unit(seed=11, n=pn)
    left = gx.InputBatches([dev_batch(k=pkeys, v=vals)])
    right = gx.InputBatches([dev_batch(bk=bkeys)])
    join = gx.ShuffledHashJoinExec("k", "bk", "right", left, right)
    agg = gx.HashAggregateExec("k", [("sum", "v")], "complete", join)
    plan = gx.SortExec(gx.SortOrder("k"), True, agg)
    got = run_plan(plan)

    op, ob = oracle.join_inner(bkeys, pkeys)
    jk, jv = pkeys[op], vals[op]
    ok, _, osum, _, _ = oracle.hash_agg(jk, jv)
    perm = oracle.sort_perm(ok)
    assert (got["k"] == ok[perm]).all()
    np.testing.assert_allclose(got["sum(v)"], osum[perm], rtol=1e-6)


def test_filter_project_pipeline():
    """filter -> project -> agg: the SURVEY §8(f).2 shape (keeps a Q1-style
    plan on-device between GPU exec nodes)."""
    n = 200_000
    keys = oracle.gen_i64(seed=20, n=n, range_=100)
    qty = oracle.gen_f64_unit(seed=21, n=n)
    price = oracle.gen_f64_unit(seed=22, n=n)
    scan = gx.InputBatches([dev_batch(k=keys, qty=qty, price=price)])
    plan = gx.HashAggregateExec(
        "k", [("sum", "revenue"), ("count", "revenue")], "complete",
        gx.ProjectExec(["k", ("revenue", "qty", "*", "price", None)],
                       gx.FilterExec("qty", "<", 0.5, scan)))
    got = run_plan(plan)
    mask = qty < 0.5
    rev = qty[mask] * price[mask]
    ok, _, osum, _, ocnt = oracle.hash_agg(keys[mask], rev)
    g, o = np.argsort(got["k"]), np.argsort(ok)
    assert (got["k"][g] == ok[o]).all()
    assert (got["count(revenue)"][g] == ocnt[o]).all()
    np.testing.assert_allclose(got["sum(revenue)"][g], osum[o], rtol=1e-6)


def test_filter_stability_and_literal_ops():
    n = 50_000
    vals = oracle.gen_i64(seed=30, n=n, range_=1000)
    from spark_amd import gpuq
    for op, fn in [("<", np.less), (">=", np.greater_equal), ("==", np.equal),
                   ("!=", np.not_equal)]:
        perm, cnt = gpuq.filter_cmp(torch.from_numpy(vals).cuda(), op, 500)
        exp = np.flatnonzero(fn(vals, 500))
        assert cnt == len(exp)
        assert (perm.cpu().numpy().astype(np.uint32) == exp.astype(np.uint32)).all()


def test_avg_aggregate():
    n = 100_000
    keys = oracle.gen_i64(seed=40, n=n, range_=333)
    vals = oracle.gen_f64_unit(seed=41, n=n)
    scan = gx.InputBatches([dev_batch(k=keys, v=vals)])
    got = run_plan(gx.HashAggregateExec("k", [("avg", "v")], "complete", scan))
    ok, _, osum, _, ocnt = oracle.hash_agg(keys, vals)
    g, o = np.argsort(got["k"]), np.argsort(ok)
    np.testing.assert_allclose(got["avg(v)"][g], osum[o] / ocnt[o], rtol=1e-6)


def test_range_scan_feed():
    # RangeExec -> GPU range + project + agg pipeline stays on-device
    plan = gx.HashAggregateExec(
        "g", [("count", "g")], "complete",
        gx.ProjectExec([("g", "id", "/", None, 97)],
                       gx.RangeExec(10_000, start=5, step=3, name="id")))
    got = run_plan(plan)
    ids = 5 + 3 * np.arange(10_000, dtype=np.int64)
    groups = ids // 97  # int64 div matches trunc toward zero for positives
    uk, cnt = np.unique(groups, return_counts=True)
    g = np.argsort(got["g"])
    assert (got["g"][g] == uk).all()
    assert (got["count(g)"][g] == cnt).all()


def test_q1_shape_multi_agg_plan():
    """The full Q1-shaped aggregate through the exec layer: several SUMs over
    different columns + AVG + COUNT in one pass."""
    n = 150_000
    keys = oracle.gen_i64(seed=50, n=n, range_=6)
    qty = oracle.gen_f64_unit(seed=51, n=n)
    price = oracle.gen_f64_unit(seed=52, n=n)
    scan = gx.InputBatches([dev_batch(k=keys, qty=qty, price=price)])
    got = run_plan(gx.HashAggregateExec(
        "k", [("sum", "qty"), ("sum", "price"), ("avg", "qty"), ("count", "price")],
        "complete", scan, capacity=64))
    ok, _, osq, _, ocnt = oracle.hash_agg(keys, qty)
    _, _, osp, _, _ = oracle.hash_agg(keys, price)
    g, o = np.argsort(got["k"]), np.argsort(ok)
    assert (got["k"][g] == ok[o]).all()
    np.testing.assert_allclose(got["sum(qty)"][g], osq[o], rtol=1e-6)
    np.testing.assert_allclose(got["sum(price)"][g], osp[o], rtol=1e-6)
    np.testing.assert_allclose(got["avg(qty)"][g], osq[o] / ocnt[o], rtol=1e-6)
    assert (got["count(price)"][g] == ocnt[o]).all()


def test_broadcast_hash_join_single_rank():
    # single rank: broadcast exchange is identity; join kernels identical
    bn, pn = 10_000, 80_000
    bkeys = oracle.gen_i64(seed=60, n=bn, range_=8_000)
    pkeys = oracle.gen_i64(seed=61, n=pn, range_=8_000)
    left = gx.InputBatches([dev_batch(lk=pkeys)])
    right = gx.BroadcastExchangeExec(gx.InputBatches([dev_batch(rk=bkeys)]))
    got = run_plan(gx.BroadcastHashJoinExec("lk", "rk", "right", left, right))
    op, ob = oracle.join_inner(bkeys, pkeys)
    assert len(got["lk"]) == len(op)
    g = np.lexsort((got["rk"], got["lk"]))
    o = np.lexsort((bkeys[ob], pkeys[op]))
    assert (got["lk"][g] == pkeys[op][o]).all()
    assert (got["rk"][g] == bkeys[ob][o]).all()


def test_aqe_get_shuffle_partitions_coalesce():
    """AQE getShuffleRDD(partitionSpecs) analog (ShuffleExchangeExec.scala:
    141, ShuffledRowRDD.scala:33 CoalescedPartitionSpec): coalescing 8 map
    partitions into 3 ranges is served straight from the partition-
    contiguous map output — no re-partition kernel runs."""
    from spark_amd import gpuq
    n, nparts = 120_000, 8
    keys = oracle.gen_i64(seed=400, n=n)
    pay = oracle.gen_i64(seed=401, n=n)
    valid = oracle.gen_i64(seed=402, n=n, range_=5) != 0
    kd, pd = torch.from_numpy(keys).cuda(), torch.from_numpy(pay).cuda()
    perm, counts = gpuq.partition_perm(kd, nparts)
    cols = {"k": gpuq.gather(kd, perm), "p": gpuq.gather(pd, perm)}
    vbits = torch.from_numpy(
        np.packbits(valid, bitorder="little")).cuda()
    vcols = {"p": gpuq.bits_to_u8(gpuq.gather_bits(vbits, perm), n)}
    splits = counts.cpu().tolist()
    offsets = [0]
    for c in splits:
        offsets.append(offsets[-1] + c)

    node = gx.GpuShuffleExchangeExec(("k",), gx.InputBatches([]))
    node._map_outputs = [(cols, vcols, offsets)]
    specs = [(0, 3), (3, 4), (4, 8)]
    batches = node.get_shuffle_partitions(specs)
    pids = oracle.partition_ids(keys, nparts)
    order = np.argsort(pids, kind="stable")
    for (lo_p, hi_p), b in zip(specs, batches):
        sel = order[(pids[order] >= lo_p) & (pids[order] < hi_p)]
        assert b.num_rows() == len(sel)
        got_k = b.column("k").cpu().numpy()
        got_p = b.column("p").cpu().numpy()
        assert (got_k == keys[sel]).all()
        gv = b.validity("p")
        got_valid = (np.unpackbits(gv.cpu().numpy(), count=len(sel),
                                   bitorder="little").astype(bool)
                     if gv is not None else np.ones(len(sel), bool))
        assert (got_valid == valid[sel]).all()
        assert (got_p[got_valid] == pay[sel][got_valid]).all()
    # total coverage: the three ranges partition the full map output
    assert sum(b.num_rows() for b in batches) == n


def test_join_streams_probe_batches():
    """the probe side is consumed batch-at-a-time (ShuffledHashJoinExec
    .doExecute streams streamedIter; the build side materializes once)."""
    bkeys = oracle.gen_i64(seed=500, n=5_000, range_=4_000)
    p1 = oracle.gen_i64(seed=501, n=8_000, range_=4_000)
    p2 = oracle.gen_i64(seed=502, n=6_000, range_=4_000)
    left = gx.InputBatches([dev_batch(lk=p1), dev_batch(lk=p2)])
    right = gx.InputBatches([dev_batch(rk=bkeys)])
    node = gx.ShuffledHashJoinExec("lk", "rk", "right", left, right)
    outs = list(gx.GpuColumnarRule().pre_columnar_transitions(node)
                .execute_columnar())
    assert len(outs) == 2
    for probe_keys, out in zip((p1, p2), outs):
        op, ob = oracle.join_inner(bkeys, probe_keys)
        assert out.num_rows() == len(op)
        g = np.lexsort((out.column("rk").cpu().numpy(),
                        out.column("lk").cpu().numpy()))
        o = np.lexsort((bkeys[ob], probe_keys[op]))
        assert (out.column("lk").cpu().numpy()[g] == probe_keys[op][o]).all()


def test_q3_plan_parity():
    """the TPC-H Q3 plan shape (bench.build_q3_plan): broadcast dimension
    join + shuffled fact join + exact scaled-decimal revenue + 3-key
    composite GROUP BY + ORDER BY revenue DESC, date — vs a numpy
    reference."""
    import bench
    rng = np.random.default_rng(17)
    nl, no, nc = 200_000, 50_000, 10_000
    li = dict(l_orderkey=rng.integers(0, no, nl),
              l_extendedprice=rng.integers(1, 10_000_00, nl),
              l_discount=rng.integers(0, 11, nl),
              l_shipdate=rng.integers(0, 2556, nl))
    od = dict(o_orderkey=np.arange(no, dtype=np.int64),
              o_custkey=rng.integers(0, nc, no),
              o_orderdate=rng.integers(0, 2556, no),
              o_shippriority=rng.integers(0, 2, no))
    cu = dict(c_custkey=np.arange(nc, dtype=np.int64),
              c_mktsegment=rng.integers(0, 5, nc))
    cut, seg = 1169, 1
    plan = bench.build_q3_plan(
        gx, dev_batch(**{k: v.astype(np.int64) for k, v in li.items()}),
        dev_batch(**{k: v.astype(np.int64) for k, v in od.items()}),
        dev_batch(**{k: v.astype(np.int64) for k, v in cu.items()}),
        cut_date=cut, segment_id=seg)
    out = list(gx.GpuColumnarRule().pre_columnar_transitions(plan)
               .execute_columnar())[0]

    # numpy reference
    seg_cust = set(np.flatnonzero(cu["c_mktsegment"] == seg).tolist())
    ok_orders = {}
    for j in range(no):
        if od["o_orderdate"][j] < cut and int(od["o_custkey"][j]) in seg_cust:
            ok_orders[j] = (int(od["o_orderdate"][j]),
                            int(od["o_shippriority"][j]))
    groups = {}
    for i in range(nl):
        okey = int(li["l_orderkey"][i])
        if li["l_shipdate"][i] > cut and okey in ok_orders:
            d, p = ok_orders[okey]
            rev = int(li["l_extendedprice"][i]) * (100 - int(li["l_discount"][i]))
            t = (okey, d, p)
            groups[t] = groups.get(t, 0) + rev
    gk = out.column("l_orderkey").cpu().numpy()
    gd = out.column("o_orderdate").cpu().numpy()
    gp = out.column("o_shippriority").cpu().numpy()
    gr = out.column("sum(revenue)").cpu().numpy()
    assert len(gk) == len(groups)
    for i in range(len(gk)):
        assert groups[(int(gk[i]), int(gd[i]), int(gp[i]))] == int(gr[i])
    # ordering: revenue DESC, then o_orderdate ASC
    for i in range(1, len(gk)):
        assert (gr[i - 1] > gr[i]
                or (gr[i - 1] == gr[i] and gd[i - 1] <= gd[i]))


def test_empty_batch_pipeline():
    """degenerate shapes: a filter that drops every row feeding sort, agg
    (grouped AND global), and join must stay well-formed end-to-end."""
    n = 10_000
    keys = oracle.gen_i64(seed=950, n=n, range_=100)
    vals = oracle.gen_f64_unit(seed=951, n=n)
    scan = gx.InputBatches([dev_batch(k=keys, v=vals)])
    dead = gx.FilterExec("k", "<", -1, scan)   # nothing passes
    out = list(gx.GpuColumnarRule().pre_columnar_transitions(
        gx.SortExec(gx.SortOrder("k"), False,
                    gx.HashAggregateExec("k", [("sum", "v"), ("count*", None)],
                                         "complete", dead)))
        .execute_columnar())[0]
    assert out.num_rows() == 0

    # global aggregate over empty input: exactly one row, COUNT 0, SUM NULL
    scan2 = gx.InputBatches([dev_batch(k=keys, v=vals)])
    dead2 = gx.FilterExec("k", "<", -1, scan2)
    out = list(gx.GpuColumnarRule().pre_columnar_transitions(
        gx.HashAggregateExec(None, [("count*", None), ("sum", "v")],
                             "complete", dead2)).execute_columnar())[0]
    assert out.num_rows() == 1
    assert int(out.column("count(1)").cpu()[0]) == 0
    sv = out.validity("sum(v)")
    assert sv is not None and int(sv.cpu()[0]) & 1 == 0  # SUM -> NULL

    # join with an empty probe side / empty build side
    empty = gx.FilterExec("k", "<", -1,
                          gx.InputBatches([dev_batch(k=keys)]))
    right = gx.InputBatches([dev_batch(rk=keys)])
    out = list(gx.GpuColumnarRule().pre_columnar_transitions(
        gx.ShuffledHashJoinExec("k", "rk", "right", empty, right))
        .execute_columnar())[0]
    assert out.num_rows() == 0
    empty2 = gx.FilterExec("rk", "<", -1,
                           gx.InputBatches([dev_batch(rk=keys)]))
    left = gx.InputBatches([dev_batch(k=keys)])
    out = list(gx.GpuColumnarRule().pre_columnar_transitions(
        gx.ShuffledHashJoinExec("k", "rk", "right", left, empty2))
        .execute_columnar())[0]
    assert out.num_rows() == 0


def test_whole_partition_sort_and_join_build_across_batches():
    """SortExec sorts the whole partition and the join build side
    materializes whole — multi-batch children (e.g. Parquet row groups)
    concatenate, never per-batch results."""
    n = 60_000
    keys = oracle.gen_i64(seed=980, n=n, range_=10_000)
    pay = oracle.gen_i64(seed=981, n=n)
    valid = oracle.gen_i64(seed=982, n=n, range_=6) != 0
    import numpy as _np
    halves = [slice(0, n // 2), slice(n // 2, n)]
    batches = [gx.ColumnarBatch(
        {"k": torch.from_numpy(keys[s]).cuda(),
         "p": torch.from_numpy(pay[s]).cuda()},
        validity={"k": torch.from_numpy(
            _np.packbits(valid[s], bitorder="little")).cuda()})
        for s in halves]
    out = list(gx.GpuColumnarRule().pre_columnar_transitions(
        gx.SortExec(gx.SortOrder("k"), False, gx.InputBatches(batches)))
        .execute_columnar())
    assert len(out) == 1 and out[0].num_rows() == n
    exp = oracle.sort_perm(keys, validity=_np.packbits(valid,
                                                       bitorder="little"))
    got_p = out[0].column("p").cpu().numpy()
    assert (got_p == pay[exp]).all()

    # join build side split across two batches
    bk = oracle.gen_i64(seed=983, n=20_000, range_=9_000)
    right = gx.InputBatches([
        dev_batch(rk=bk[:12_000]), dev_batch(rk=bk[12_000:])])
    left = gx.InputBatches([dev_batch(k=keys)])
    j = list(gx.GpuColumnarRule().pre_columnar_transitions(
        gx.ShuffledHashJoinExec("k", "rk", "right", left, right))
        .execute_columnar())[0]
    op, ob = oracle.join_inner(bk, keys)
    assert j.num_rows() == len(op)
