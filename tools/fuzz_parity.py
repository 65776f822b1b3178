"""Randomized GPU-vs-oracle parity fuzz: random shapes, ranges, descs,
validity patterns across sort/partition/agg/join. Seeded per iteration and
bounded by FUZZ_SECONDS (default 240)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch
import oracle
from spark_amd import gpuq as gq

BUDGET = float(os.environ.get("FUZZ_SECONDS", 240))
rng = np.random.default_rng(int(os.environ.get("FUZZ_SEED", 0)))
t_end = time.time() + BUDGET
it = 0
fails = 0

def dev(a):
    return torch.from_numpy(np.ascontiguousarray(a)).cuda()

while time.time() < t_end:
    it += 1
    op = rng.choice(["sort_i64", "sort_f64", "sort_nulls", "partition",
                     "agg", "join", "minmax_agg", "composite_agg",
                     "typed_join", "partition_multi"])
    n = int(rng.integers(1, 3_000_000))
    seed = int(rng.integers(0, 2**31))
    try:
        if op == "sort_i64":
            r = int(rng.choice([0, 1, 10, 1000, 10**6, 2**62]))
            keys = oracle.gen_i64(seed, n, range_=r)
            desc = bool(rng.integers(2))
            perm, skeys = gq.sort_perm(dev(keys), desc=desc)
            exp = oracle.sort_perm(keys, desc=desc)
            assert (perm.cpu().numpy().astype(np.uint32) == exp.astype(np.uint32)).all()
            assert (skeys.cpu().numpy() == keys[exp]).all()
        elif op == "sort_f64":
            keys = oracle.gen_f64_unit(seed, n)
            if n > 10:
                keys[:: max(1, n // 50)] = rng.choice([np.nan, np.inf, -np.inf, -0.0, 0.0])
            desc = bool(rng.integers(2))
            perm, _ = gq.sort_perm(dev(keys), desc=desc)
            exp = oracle.sort_perm(keys, desc=desc)
            assert (perm.cpu().numpy().astype(np.uint32) == exp.astype(np.uint32)).all()
        elif op == "sort_nulls":
            keys = oracle.gen_i64(seed, n, range_=max(1, n // 10))
            valid = rng.random(n) > rng.random() * 0.9
            vb = np.packbits(valid, bitorder="little")
            desc = bool(rng.integers(2))
            perm, _ = gq.sort_perm(dev(keys), desc=desc,
                                   key_validity=torch.from_numpy(vb).cuda())
            exp = oracle.sort_perm(keys, desc=desc, validity=vb)
            assert (perm.cpu().numpy().astype(np.uint32) == exp.astype(np.uint32)).all()
        elif op == "partition":
            nparts = int(rng.choice([1, 2, 7, 8, 200, 256, 1000, 65536]))
            keys = oracle.gen_i64(seed, n)
            perm, counts = gq.partition_perm(dev(keys), nparts)
            pids = oracle.partition_ids(keys, nparts)
            assert (counts.cpu().numpy() == np.bincount(pids, minlength=nparts)).all()
            assert (perm.cpu().numpy().astype(np.uint32)
                    == np.argsort(pids, kind="stable").astype(np.uint32)).all()
        elif op == "agg":
            groups = int(rng.integers(1, max(2, n // 2)))
            keys = oracle.gen_i64(seed, n, range_=groups)
            if rng.integers(4) == 0:
                keys[:: max(1, n // 17)] = -1
            vals = oracle.gen_f64_unit(seed + 1, n)
            cap = 1 << max(4, int(groups * 2 + 2).bit_length())
            gk, gkv, gs, gsv, gc = (t.cpu().numpy() for t in
                                    gq.hash_agg(dev(keys), dev(vals), cap))
            ok, okv, osum, osv, ocnt = oracle.hash_agg(keys, vals)
            g, o = np.lexsort((gk, gkv)), np.lexsort((ok, okv))
            assert (gk[g] == ok[o]).all() and (gc[g] == ocnt[o]).all()
            np.testing.assert_allclose(gs[g], osum[o], rtol=1e-6)
        elif op == "join":
            bn = int(rng.integers(1, max(2, n)))
            r = max(1, int(rng.integers(1, 2 * n)))
            bkeys = oracle.gen_i64(seed, bn, range_=r)
            pkeys = oracle.gen_i64(seed + 1, n, range_=r)
            cap = 1 << max(4, int(bn * 2 + 2).bit_length())
            ws = gq.join_build(dev(bkeys), cap)
            op_o, ob_o = oracle.join_inner(bkeys, pkeys)
            gp, gb, nm = gq.join_probe(dev(pkeys), ws, cap, bn, len(op_o) + 64)
            assert nm == len(op_o)
            g = np.lexsort((gb.cpu().numpy(), gp.cpu().numpy()))
            o = np.lexsort((ob_o, op_o))
            assert (gp.cpu().numpy()[g] == op_o[o]).all()
            assert (gb.cpu().numpy()[g] == ob_o[o]).all()
        elif op == "minmax_agg":
            groups = int(rng.integers(1, max(2, n // 2)))
            keys = oracle.gen_i64(seed, n, range_=groups)
            vals = oracle.gen_i64(seed + 1, n)
            valid = rng.random(n) > rng.random() * 0.8
            vb = torch.from_numpy(np.packbits(valid, bitorder="little")).cuda()
            cap = 1 << max(4, int(groups * 2 + 2).bit_length())
            ok_, okv_, accs = gq.hash_agg_multi(
                dev(keys), [("min", dev(vals), vb), ("max", dev(vals), vb),
                            ("count", dev(vals), vb)], cap)
            gk = ok_.cpu().numpy()
            gmin, gmax, gcnt = (t.cpu().numpy() for t in accs)
            ref = {}
            for i in range(n):
                ref.setdefault(int(keys[i]), []).append(i)
            assert len(gk) == len(ref)
            for j, k in enumerate(gk):
                vv = [vals[i] for i in ref[int(k)] if valid[i]]
                assert gcnt[j] == len(vv)
                if vv:
                    assert gmin[j] == min(vv) and gmax[j] == max(vv)
        elif op == "composite_agg":
            g1 = int(rng.integers(1, 100))
            g2 = int(rng.integers(1, 100))
            k1 = oracle.gen_i64(seed, n, range_=g1)
            k2 = oracle.gen_i64(seed + 1, n, range_=g2)
            vals = oracle.gen_f64_unit(seed + 2, n)
            cap = 1 << max(6, int(g1 * g2 * 2 + 2).bit_length())
            okeys, kmask, accs = gq.hash_agg_keys(
                [dev(k1), dev(k2)], [("sum", dev(vals), None), ("count*",)],
                cap)
            ref = {}
            for i in range(n):
                t_ = (int(k1[i]), int(k2[i]))
                ref.setdefault(t_, [0.0, 0])
                ref[t_][0] += vals[i]
                ref[t_][1] += 1
            a1, a2 = okeys[0].cpu().numpy(), okeys[1].cpu().numpy()
            gs_, gc_ = accs[0].cpu().numpy(), accs[1].cpu().numpy()
            assert len(a1) == len(ref)
            for j in range(len(a1)):
                s, c = ref[(int(a1[j]), int(a2[j]))]
                assert gc_[j] == c
                np.testing.assert_allclose(gs_[j], s, rtol=1e-6)
        elif op == "typed_join":
            jt = int(rng.choice([1, 2, 3]))
            bn = int(rng.integers(1, max(2, n)))
            r = max(1, int(rng.integers(1, 2 * n)))
            bkeys = oracle.gen_i64(seed, bn, range_=r)
            pkeys = oracle.gen_i64(seed + 1, n, range_=r)
            cap = 1 << max(4, int(bn * 2 + 2).bit_length())
            ws = gq.join_build(dev(bkeys), cap)
            bset = set(bkeys.tolist())
            from collections import Counter
            bc = Counter(bkeys.tolist())
            if jt == 1:       # outer: matches + 1 per unmatched probe row
                exp_n = sum(bc.get(int(k), 0) or 1 for k in pkeys)
            elif jt == 2:     # semi
                exp_n = sum(1 for k in pkeys if int(k) in bset)
            else:             # anti
                exp_n = sum(1 for k in pkeys if int(k) not in bset)
            if exp_n > 20_000_000:
                continue   # skip pathological cross products (alloc bound)
            gp, gb, nm = gq.join_probe(dev(pkeys), ws, cap, bn,
                                       exp_n + 64, join_type=jt)
            assert nm == exp_n
        else:  # partition_multi
            nparts = int(rng.choice([1, 2, 8, 200]))
            k1 = oracle.gen_i64(seed, n)
            k2 = oracle.gen_i64(seed + 1, n, range_=100)
            perm, counts = gq.partition_perm_multi([dev(k1), dev(k2)], nparts)
            ones = np.packbits(np.ones(n, dtype=bool), bitorder="little")
            pids = oracle.partition_ids_multi(
                [k1, k2], nparts, validity=np.stack([ones, ones]))
            assert (counts.cpu().numpy()
                    == np.bincount(pids, minlength=nparts)).all()
            assert (perm.cpu().numpy().astype(np.uint32)
                    == np.argsort(pids, kind="stable").astype(np.uint32)).all()
    except AssertionError:
        fails += 1
        print(f"FAIL it={it} op={op} n={n} seed={seed}", flush=True)
        if fails > 3:
            raise
print(f"fuzz done: {it} iterations, {fails} failures", flush=True)
assert fails == 0
