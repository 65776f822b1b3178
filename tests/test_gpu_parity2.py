"""GPU parity for the round-2 kernels: MIN/MAX aggregates, NULL-skipping
SUMs, composite-key aggregation, multi-column partition ids, validity
bitmap utilities, and the narrow-key pack rule — each against the oracle /
an independent numpy reference on the same seeded inputs."""
import numpy as np
import pytest

import oracle

torch = pytest.importorskip("torch")
pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def gq():
    from spark_amd import gpuq
    assert torch.cuda.is_available()
    return gpuq


def to_dev(a: np.ndarray):
    return torch.from_numpy(np.ascontiguousarray(a)).cuda()


def pack_validity(valid_bool: np.ndarray):
    return torch.from_numpy(np.packbits(valid_bool, bitorder="little")).cuda()


def ref_groupby(keys, valid=None):
    """-> dict key(or None) -> row indices (NULLs one group)."""
    groups = {}
    for i, k in enumerate(keys):
        kk = None if (valid is not None and not valid[i]) else int(k)
        groups.setdefault(kk, []).append(i)
    return groups


# ---------- MIN/MAX + NULL-skipping SUM in the multi-agg ----------

@pytest.mark.parametrize("dtype", ["i64", "f64"])
def test_minmax_agg_parity(gq, dtype):
    n, ngroups = 300_000, 512
    keys = oracle.gen_i64(seed=1, n=n, range_=ngroups)
    if dtype == "i64":
        vals = oracle.gen_i64(seed=2, n=n)
    else:
        vals = oracle.gen_f64_unit(seed=2, n=n) * 2 - 1
        vals[::97] = -0.0
        vals[::89] = np.inf
        vals[::83] = -np.inf
        vals[::101] = np.nan
    valid = oracle.gen_i64(seed=3, n=n, range_=5) != 0
    ok, okv, accs = gq.hash_agg_multi(
        to_dev(keys),
        [("min", to_dev(vals), pack_validity(valid)),
         ("max", to_dev(vals), pack_validity(valid)),
         ("count", to_dev(vals), pack_validity(valid))],
        1 << 12)
    gk = ok.cpu().numpy()
    gmin, gmax, gcnt = (t.cpu().numpy() for t in accs)
    groups = ref_groupby(keys)
    assert len(gk) == len(groups)
    for i, k in enumerate(gk):
        rows = groups[int(k)]
        vv = [vals[r] for r in rows if valid[r]]
        assert gcnt[i] == len(vv)
        if not vv:
            continue  # all-NULL group: host layer maps count==0 to NULL
        if dtype == "i64":
            assert gmin[i] == min(vv) and gmax[i] == max(vv)
        else:
            # Java Double.compare order: NaN greatest, -0.0 < 0.0
            enc = [oracle.prefix_double(float(v)) for v in vv]
            lo = vv[int(np.argmin(enc))]
            hi = vv[int(np.argmax(enc))]
            assert (np.isnan(gmin[i]) and np.isnan(lo)) or gmin[i] == lo
            assert (np.isnan(gmax[i]) and np.isnan(hi)) or gmax[i] == hi


def test_nullskip_sum_parity(gq):
    n = 200_000
    keys = oracle.gen_i64(seed=5, n=n, range_=100)
    vals = oracle.gen_f64_unit(seed=6, n=n)
    valid = oracle.gen_i64(seed=7, n=n, range_=3) != 0
    ok, okv, accs = gq.hash_agg_multi(
        to_dev(keys),
        [("sum", to_dev(vals), pack_validity(valid)),
         ("count", to_dev(vals), pack_validity(valid))],
        1 << 10)
    gk = ok.cpu().numpy()
    gsum, gcnt = accs[0].cpu().numpy(), accs[1].cpu().numpy()
    groups = ref_groupby(keys)
    for i, k in enumerate(gk):
        rows = groups[int(k)]
        vv = [vals[r] for r in rows if valid[r]]
        assert gcnt[i] == len(vv)
        np.testing.assert_allclose(gsum[i], sum(vv), rtol=1e-6, atol=1e-12)


def test_nullskip_sum_i64_wrapping(gq):
    # SUM(int64) merges as wrapping two's-complement adds (Sum.scala
    # LongType, non-ansi) — NULLs skipped
    keys = np.zeros(4, dtype=np.int64)
    vals = np.array([(1 << 62), (1 << 62), (1 << 62), 12345], dtype=np.int64)
    valid = np.array([True, True, True, False])
    ok, okv, accs = gq.hash_agg_multi(
        to_dev(keys), [("sum", to_dev(vals), pack_validity(valid))], 1 << 4)
    expected = ((3 * (1 << 62)) + (1 << 63)) % (1 << 64) - (1 << 63)
    assert accs[0].cpu().numpy()[0] == expected


# ---------- composite-key aggregation ----------

@pytest.mark.parametrize("nkeys", [2, 3])
def test_composite_key_agg_parity(gq, nkeys):
    n = 250_000
    kc = [oracle.gen_i64(seed=10 + c, n=n, range_=50) for c in range(nkeys)]
    vals = oracle.gen_f64_unit(seed=20, n=n)
    kvalid = [oracle.gen_i64(seed=30 + c, n=n, range_=8) != 0
              for c in range(nkeys)]
    okeys, kmask, accs = gq.hash_agg_keys(
        [to_dev(k) for k in kc],
        [("sum", to_dev(vals), None), ("count*",)],
        1 << (14 if nkeys == 2 else 19),   # ~51^nkeys distinct tuples
        key_validities=[pack_validity(v) for v in kvalid])
    gk = [t.cpu().numpy() for t in okeys]
    gm = kmask.cpu().numpy()
    gsum, gcnt = accs[0].cpu().numpy(), accs[1].cpu().numpy()
    ref = {}
    for i in range(n):
        t = tuple(int(kc[c][i]) if kvalid[c][i] else None
                  for c in range(nkeys))
        ref.setdefault(t, []).append(i)
    assert len(gk[0]) == len(ref)
    for g in range(len(gk[0])):
        t = tuple(int(gk[c][g]) if (gm[g] >> c) & 1 else None
                  for c in range(nkeys))
        rows = ref[t]
        assert gcnt[g] == len(rows)
        np.testing.assert_allclose(gsum[g], sum(vals[r] for r in rows),
                                   rtol=1e-6)


def test_composite_key_agg_sentinel_keys(gq):
    # keys spanning the sentinel-looking values (-1, 0, INT64_MIN/MAX) and
    # NULLs — the sig-claim scheme has no reserved key values
    k1 = np.array([-1, -1, 0, (1 << 63) - 1, -(1 << 63), -1, 0],
                  dtype=np.int64)
    k2 = np.array([-1, -1, -1, 5, 5, 0, -1], dtype=np.int64)
    v1 = np.array([True, True, True, True, True, True, False])
    vals = np.arange(7, dtype=np.float64)
    okeys, kmask, accs = gq.hash_agg_keys(
        [to_dev(k1), to_dev(k2)],
        [("sum", to_dev(vals), None), ("count*",)],
        1 << 6, key_validities=[pack_validity(v1), None])
    ref = {}
    for i in range(7):
        t = (int(k1[i]) if v1[i] else None, int(k2[i]))
        ref.setdefault(t, 0.0)
        ref[t] += vals[i]
    gk1, gk2 = okeys[0].cpu().numpy(), okeys[1].cpu().numpy()
    gm = kmask.cpu().numpy()
    gsum = accs[0].cpu().numpy()
    assert len(gk1) == len(ref)
    for g in range(len(gk1)):
        t = (int(gk1[g]) if gm[g] & 1 else None, int(gk2[g]))
        np.testing.assert_allclose(gsum[g], ref[t], rtol=1e-12)


# ---------- multi-column partition ids ----------

def test_partition_perm_multi_parity(gq):
    n = 200_000
    nparts = 8
    k1 = oracle.gen_i64(seed=40, n=n)
    k2 = oracle.gen_i64(seed=41, n=n, range_=1000)
    valid1 = oracle.gen_i64(seed=42, n=n, range_=6) != 0
    perm, counts = gq.partition_perm_multi(
        [to_dev(k1), to_dev(k2)], nparts,
        key_validities=[pack_validity(valid1), None])
    exp_pids = oracle.partition_ids_multi(
        [k1, k2], nparts,
        validity=np.stack([np.packbits(valid1, bitorder="little"),
                           np.packbits(np.ones(n, dtype=bool),
                                       bitorder="little")]))
    exp_perm = np.argsort(exp_pids, kind="stable")
    assert (perm.cpu().numpy().astype(np.int64) == exp_perm).all()
    exp_counts = np.bincount(exp_pids, minlength=nparts)
    assert (counts.cpu().numpy() == exp_counts).all()


# ---------- validity bitmap utilities ----------

def test_gather_bits_roundtrip(gq):
    n = 100_003
    valid = oracle.gen_i64(seed=50, n=n, range_=3) != 0
    perm = np.argsort(oracle.gen_i64(seed=51, n=n), kind="stable") \
        .astype(np.uint32)
    out = gq.gather_bits(pack_validity(valid),
                         to_dev(perm.astype(np.int32)))
    got = np.unpackbits(out.cpu().numpy(), count=n, bitorder="little") \
        .astype(bool)
    assert (got == valid[perm]).all()


def test_bits_u8_roundtrip(gq):
    n = 77_77
    valid = oracle.gen_i64(seed=60, n=n, range_=2) != 0
    bits = pack_validity(valid)
    u8 = gq.bits_to_u8(bits, n)
    assert (u8.cpu().numpy().astype(bool) == valid).all()
    back = gq.u8_to_bits(u8)
    got = np.unpackbits(back.cpu().numpy(), count=n, bitorder="little") \
        .astype(bool)
    assert (got == valid).all()


def test_nonzero_and_maskbit_bits(gq):
    n = 4099
    cnt = oracle.gen_i64(seed=70, n=n, range_=3)
    bits = gq.nonzero_to_bits(to_dev(cnt))
    got = np.unpackbits(bits.cpu().numpy(), count=n, bitorder="little") \
        .astype(bool)
    assert (got == (cnt != 0)).all()
    mask = oracle.gen_i64(seed=71, n=n, range_=256).astype(np.uint8)
    for bit in (0, 1, 3):
        mb = gq.maskbit_to_bits(to_dev(mask), bit)
        got = np.unpackbits(mb.cpu().numpy(), count=n, bitorder="little") \
            .astype(bool)
        assert (got == ((mask >> bit) & 1).astype(bool)).all()


# ---------- minmax reduction + pack rule ----------

def test_minmax_i64(gq):
    n = 500_000
    vals = oracle.gen_i64(seed=80, n=n)
    valid = oracle.gen_i64(seed=81, n=n, range_=4) != 0
    mn, mx, cnt = gq.minmax_i64(to_dev(vals), validity=pack_validity(valid))
    vv = vals[valid]
    assert cnt == len(vv) and mn == vv.min() and mx == vv.max()
    mn2, mx2, cnt2 = gq.minmax_i64(to_dev(vals))
    assert cnt2 == n and mn2 == vals.min() and mx2 == vals.max()
    _, _, zero = gq.minmax_i64(to_dev(vals),
                               validity=pack_validity(np.zeros(n, dtype=bool)))
    assert zero == 0


def test_pack2_unpack2_roundtrip(gq):
    n = 300_000
    a = oracle.gen_i64(seed=90, n=n, range_=1000) - 17
    b = oracle.gen_i64(seed=91, n=n, range_=5000) + 3
    shift = 13  # 5000 < 2^13
    packed = gq.pack2_i64(to_dev(a), to_dev(b), int(a.min()), int(b.min()),
                          shift)
    exp = ((a - a.min()) << shift) | (b - b.min())
    assert (packed.cpu().numpy() == exp).all()
    ua, ub = gq.unpack2_i64(packed, int(a.min()), int(b.min()), shift)
    assert (ua.cpu().numpy() == a).all() and (ub.cpu().numpy() == b).all()


# ---------- partial -> final merge through the exec mirror ----------

def test_partial_final_merge_exec(gq):
    """two partial batches merged by a final-mode node: counts merge as
    exact int64 sums, sums/min/max merge NULL-aware (the AggUtils
    Partial/Final split, exercised without an exchange)."""
    from spark_amd import exec as gx
    n = 120_000
    keys = oracle.gen_i64(seed=100, n=n, range_=200)
    vals = oracle.gen_f64_unit(seed=101, n=n)
    valid = oracle.gen_i64(seed=102, n=n, range_=4) != 0
    half = n // 2
    batches = []
    for lo, hi in ((0, half), (half, n)):
        batches.append(gx.ColumnarBatch(
            {"k": to_dev(keys[lo:hi]), "v": to_dev(vals[lo:hi])},
            validity={"v": pack_validity(valid[lo:hi])}))
    aggs = [("sum", "v"), ("count", "v"), ("avg", "v"),
            ("min", "v"), ("max", "v"), ("count*", None)]
    # two partial nodes = two RANKS, each aggregating its own partition
    # batches (the whole-partition contract: one output per partition)
    partials = []
    for bat in batches:
        node = gx.HashAggregateExec("k", aggs, "partial",
                                    gx.InputBatches([bat]))
        outs = list(gx.GpuColumnarRule().pre_columnar_transitions(node)
                    .execute_columnar())
        assert len(outs) == 1
        partials.append(outs[0])
    # concat partial outputs into one batch (what the exchange would yield)
    cols, validity = {}, {}
    names = list(partials[0].columns().keys())
    for nme in names:
        cols[nme] = torch.cat([p.column(nme) for p in partials])
        vs = [p.validity(nme) for p in partials]
        if any(v is not None for v in vs):
            cat = []
            for p, v in zip(partials, vs):
                nn = p.num_rows()
                if v is None:
                    cat.append(np.ones(nn, dtype=bool))
                else:
                    cat.append(np.unpackbits(v.cpu().numpy(), count=nn,
                                             bitorder="little").astype(bool))
            validity[nme] = pack_validity(np.concatenate(cat))
    merged = gx.ColumnarBatch(cols, validity=validity or None)
    final_node = gx.HashAggregateExec("k", aggs, "final",
                                      gx.InputBatches([merged]))
    out = list(gx.GpuColumnarRule().pre_columnar_transitions(final_node)
               .execute_columnar())[0]
    gk = out.column("k").cpu().numpy()
    groups = ref_groupby(keys)
    assert len(gk) == len(groups)
    gsum = out.column("sum(v)").cpu().numpy()
    gcnt = out.column("count(v)").cpu().numpy()
    gavg = out.column("avg(v)").cpu().numpy()
    gmin = out.column("min(v)").cpu().numpy()
    gmax = out.column("max(v)").cpu().numpy()
    gstar = out.column("count(1)").cpu().numpy()
    for i, k in enumerate(gk):
        rows = groups[int(k)]
        vv = [vals[r] for r in rows if valid[r]]
        assert gstar[i] == len(rows)
        assert gcnt[i] == len(vv)
        if vv:
            np.testing.assert_allclose(gsum[i], sum(vv), rtol=1e-6)
            np.testing.assert_allclose(gavg[i], sum(vv) / len(vv), rtol=1e-6)
            assert gmin[i] == min(vv) and gmax[i] == max(vv)


# ---------- typed joins (outer / semi / anti over the probe side) ----------

@pytest.mark.parametrize("jt_name", ["left_outer", "left_semi", "left_anti", "full_outer"])
def test_typed_join_parity(gq, jt_name):
    """probe-side outer/semi/anti against a python reference, with NULL
    probe keys, duplicate build chains and unmatched rows
    (ShuffledHashJoinExec.scala joinType dispatch)."""
    from spark_amd import exec as gx
    bn, pn = 40_000, 70_000
    bkeys = oracle.gen_i64(seed=201, n=bn, range_=30_000)
    bpay = oracle.gen_i64(seed=202, n=bn)
    pkeys = oracle.gen_i64(seed=203, n=pn, range_=60_000)  # ~half unmatched
    ppay = oracle.gen_i64(seed=204, n=pn)
    pvalid = oracle.gen_i64(seed=205, n=pn, range_=10) != 0
    left = gx.InputBatches([gx.ColumnarBatch(
        {"lk": to_dev(pkeys), "lp": to_dev(ppay)},
        validity={"lk": pack_validity(pvalid)})])
    right = gx.InputBatches([gx.ColumnarBatch(
        {"rk": to_dev(bkeys), "rp": to_dev(bpay)})])
    node = gx.ShuffledHashJoinExec("lk", "rk", "right", left, right,
                                   join_type=jt_name)
    out = list(gx.GpuColumnarRule().pre_columnar_transitions(node)
               .execute_columnar())[0]

    import collections
    buckets = collections.defaultdict(list)
    for j, v in enumerate(bkeys):
        buckets[int(v)].append(j)
    exp = []   # (probe_rid, build_rid or None)
    for i in range(pn):
        hits = buckets.get(int(pkeys[i]), []) if pvalid[i] else []
        if jt_name in ("left_outer", "full_outer"):
            exp += [(i, j) for j in hits] if hits else [(i, None)]
        elif jt_name == "left_semi":
            exp += [(i, None)] if hits else []
        else:
            exp += [] if hits else [(i, None)]
    if jt_name == "full_outer":
        matched_b = {j for _, j in exp if j is not None}
        exp += [(None, j) for j in range(bn) if j not in matched_b]
    got_lk = out.column("lk").cpu().numpy()
    got_lp = out.column("lp").cpu().numpy()
    lkv = out.validity("lk")
    got_lkv = (np.unpackbits(lkv.cpu().numpy(), count=len(got_lk),
                             bitorder="little").astype(bool)
               if lkv is not None else np.ones(len(got_lk), bool))
    assert len(got_lk) == len(exp)
    # compare as multisets of full rows
    def row(i, j):
        if i is None:
            return (None, None, int(bkeys[j]), int(bpay[j]))
        lk = int(pkeys[i]) if pvalid[i] else None
        if jt_name in ("left_outer", "full_outer"):
            rp = int(bpay[j]) if j is not None else None
            rk = int(bkeys[j]) if j is not None else None
            return (lk, int(ppay[i]), rk, rp)
        return (lk, int(ppay[i]))
    expected = sorted((row(i, j) for i, j in exp),
                      key=lambda r: tuple((x is None, x or 0) for x in r))
    if jt_name in ("left_outer", "full_outer"):
        got_rk = out.column("rk").cpu().numpy()
        got_rp = out.column("rp").cpu().numpy()
        rkv = out.validity("rk")
        got_rkv = np.unpackbits(rkv.cpu().numpy(), count=len(got_lk),
                                bitorder="little").astype(bool)
        lpv = out.validity("lp")
        got_lpv = (np.unpackbits(lpv.cpu().numpy(), count=len(got_lk),
                                 bitorder="little").astype(bool)
                   if lpv is not None else np.ones(len(got_lk), bool))
        got = [(int(got_lk[i]) if got_lkv[i] else None,
                int(got_lp[i]) if got_lpv[i] else None,
                int(got_rk[i]) if got_rkv[i] else None,
                int(got_rp[i]) if got_rkv[i] else None)
               for i in range(len(got_lk))]
    else:
        got = [(int(got_lk[i]) if got_lkv[i] else None, int(got_lp[i]))
               for i in range(len(got_lk))]
    got = sorted(got, key=lambda r: tuple((x is None, x or 0) for x in r))
    assert got == expected


def test_filter_fractional_literal_on_int64(gq):
    """Spark compares int columns against fractional literals via
    cast-to-double; the engine rewrites to equivalent integer predicates
    (k < 0.5 must keep k == 0 — the truncation bug class)."""
    n = 100_000
    vals = oracle.gen_i64(seed=900, n=n, range_=7) - 3   # -3..3
    dv = to_dev(vals)
    for op, lit in [("<", 0.5), ("<=", 0.5), (">", 0.5), (">=", 0.5),
                    ("<", -1.5), (">", -1.5), ("==", 0.5), ("!=", 0.5),
                    ("<", 1e300), (">", 1e300)]:
        perm, cnt = gq.filter_cmp(dv, op, lit)
        ref = {"<": vals < lit, "<=": vals <= lit, ">": vals > lit,
               ">=": vals >= lit, "==": vals == lit, "!=": vals != lit}[op]
        exp = np.flatnonzero(ref)
        assert cnt == len(exp), (op, lit, cnt, len(exp))
        assert (perm.cpu().numpy().astype(np.int64) == exp).all(), (op, lit)


def test_null_aware_anti_join(gq):
    """NOT IN rewrite (BroadcastHashJoinExec.scala isNullAwareAntiJoin):
    empty build -> all probe rows; any NULL build key -> empty; otherwise
    anti with NULL probe keys filtered (NULL NOT IN (...) is unknown)."""
    from spark_amd import exec as gx
    pn = 50_000
    pkeys = oracle.gen_i64(seed=700, n=pn, range_=1000)
    pvalid = oracle.gen_i64(seed=701, n=pn, range_=8) != 0

    def probe_batches():
        return gx.InputBatches([gx.ColumnarBatch(
            {"lk": to_dev(pkeys)}, validity={"lk": pack_validity(pvalid)})])

    def run(bkeys, bvalid=None):
        right = gx.InputBatches([gx.ColumnarBatch(
            {"rk": to_dev(bkeys)},
            validity={"rk": pack_validity(bvalid)} if bvalid is not None
            else None)])
        node = gx.GpuShuffledHashJoinExec(
            "lk", "rk", "right", probe_batches(), right,
            join_type="left_anti", null_aware_anti=True)
        return list(node.execute_columnar())[0]

    # empty build: every probe row (NULLs included)
    out = run(np.empty(0, dtype=np.int64))
    assert out.num_rows() == pn
    # build containing a NULL key: empty result
    bk = oracle.gen_i64(seed=702, n=100, range_=500)
    bv = np.ones(100, dtype=bool); bv[17] = False
    out = run(bk, bv)
    assert out.num_rows() == 0
    # plain: probe keys not in the build set, NULL probe keys filtered
    out = run(bk)
    bset = set(bk.tolist())
    exp = sorted(int(pkeys[i]) for i in range(pn)
                 if pvalid[i] and int(pkeys[i]) not in bset)
    got = sorted(out.column("lk").cpu().numpy().tolist())
    assert got == exp


def test_agg_accumulates_across_batches(gq):
    """one hash table per PARTITION: multiple child batches accumulate and
    emit a single result (TungstenAggregationIterator per-task contract)."""
    from spark_amd import exec as gx
    n = 90_000
    keys = oracle.gen_i64(seed=970, n=n, range_=500)
    vals = oracle.gen_f64_unit(seed=971, n=n)
    thirds = [slice(0, n // 3), slice(n // 3, 2 * n // 3), slice(2 * n // 3, n)]
    batches = [gx.ColumnarBatch({"k": to_dev(keys[s]), "v": to_dev(vals[s])})
               for s in thirds]
    node = gx.HashAggregateExec("k", [("sum", "v"), ("count*", None)],
                                "complete", gx.InputBatches(batches))
    outs = list(gx.GpuColumnarRule().pre_columnar_transitions(node)
                .execute_columnar())
    assert len(outs) == 1
    out = outs[0]
    ok, _, osum, _, _ = oracle.hash_agg(keys, vals)
    assert out.num_rows() == len(ok)
    gk = out.column("k").cpu().numpy()
    gs = out.column("sum(v)").cpu().numpy()
    gc = out.column("count(1)").cpu().numpy()
    g, o = np.argsort(gk), np.argsort(ok)
    assert (gk[g] == ok[o]).all()
    np.testing.assert_allclose(gs[g], osum[o], rtol=1e-6)
    ref_cnt = np.bincount(keys, minlength=500)
    assert (gc[g] == ref_cnt[ok[o]]).all()

    # composite keys across batches too
    k2 = oracle.gen_i64(seed=972, n=n, range_=7)
    b2 = [gx.ColumnarBatch({"a": to_dev(keys[s]), "b": to_dev(k2[s]),
                            "v": to_dev(vals[s])}) for s in thirds]
    node2 = gx.HashAggregateExec(("a", "b"), [("count*", None)], "complete",
                                 gx.InputBatches(b2))
    out2 = list(gx.GpuColumnarRule().pre_columnar_transitions(node2)
                .execute_columnar())[0]
    ref = {}
    for i in range(n):
        ref[(int(keys[i]), int(k2[i]))] = ref.get((int(keys[i]), int(k2[i])), 0) + 1
    assert out2.num_rows() == len(ref)
    ga = out2.column("a").cpu().numpy()
    gb = out2.column("b").cpu().numpy()
    gc2 = out2.column("count(1)").cpu().numpy()
    for i in range(len(ga)):
        assert ref[(int(ga[i]), int(gb[i]))] == int(gc2[i])


def test_hash_agg_mid_cardinality_routing(gq):
    """the auto-route band (cap 2^16..2^22, n >= 2^24) must take the
    partitioned kernel and stay parity-green; outside the band the direct
    table runs. Both against the oracle."""
    n = 1 << 24
    groups = 60_000
    keys = oracle.gen_i64(seed=990, n=n, range_=groups)
    vals = oracle.gen_f64_unit(seed=991, n=n)
    cap = 1 << 17   # in band
    gq.kernel_stats_reset()
    gq.profiling(True)
    gk, gkv, gs, gsv, gc = (t.cpu().numpy() for t in
                            gq.hash_agg(to_dev(keys), to_dev(vals), cap))
    _, scatter_calls = gq.kernel_stats("pagg_scatter")
    assert scatter_calls > 0, "band input did not route to partitioned"
    ok, okv, osum, osv, ocnt = oracle.hash_agg(keys, vals)
    g, o = np.argsort(gk), np.argsort(ok)
    assert (gk[g] == ok[o]).all() and (gc[g] == ocnt[o]).all()
    np.testing.assert_allclose(gs[g], osum[o], rtol=1e-6)
    gq.profiling(False)
