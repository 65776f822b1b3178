"""Diagnostic: how do agg/join kernel times scale with table footprint?
(If small tables are fast, locality-bucketing pays; if flat, we are
atomic/line-throughput bound.)"""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from spark_amd import gpuq as gq

gq.profiling(True)
n = 1_000_000_000

for groups in [1_000, 100_000, 10_000_000, 100_000_000]:
    keys = gq.gen_i64(seed=1, n=n, range_=groups)
    vals = gq.gen_f64_unit(seed=2, n=n)
    cap = 1 << max(10, (groups * 2 - 1).bit_length())
    ws = gq.agg_workspace(cap)
    for _ in range(2):
        gq.hash_agg(keys, vals, cap, workspace=ws, max_groups=groups + 2, ops=gq.AGG_SUM)
    gq.kernel_stats_reset()
    gq.hash_agg(keys, vals, cap, workspace=ws, max_groups=groups + 2, ops=gq.AGG_SUM)
    ms, _ = gq.kernel_stats("agg_build")
    print(f"agg groups={groups:>11,} table={cap*24/2**20:8.1f}MB build={ms:7.2f}ms", flush=True)
    del keys, vals, ws
    torch.cuda.empty_cache()

pn = 500_000_000
pkeys = gq.gen_i64(seed=3, n=pn, range_=pn)
for bn in [1_000_000, 30_000_000, 500_000_000]:
    bkeys = gq.gen_i64(seed=4, n=bn, range_=pn)  # same keyspace
    cap = 1 << (int(bn * 1.6) - 1).bit_length()
    ws = gq.join_build(bkeys, cap)
    out_cap = int(pn * 1.3) + 65536
    for _ in range(2):
        gq.join_probe(pkeys, ws, cap, bn, out_cap)
    gq.kernel_stats_reset()
    gq.join_probe(pkeys, ws, cap, bn, out_cap)
    ms, _ = gq.kernel_stats("join_probe")
    print(f"join bn={bn:>11,} table={cap*16/2**20:8.1f}MB probe={ms:7.2f}ms", flush=True)
    del bkeys, ws
    torch.cuda.empty_cache()

# LDS pre-agg A/B at low cardinality (re-run section)
if os.environ.get("DIAG_LDS_AGG"):
    n = 1_000_000_000
    for groups in [4, 1_000, 500_000]:
        keys = gq.gen_i64(seed=1, n=n, range_=groups)
        vals = gq.gen_f64_unit(seed=2, n=n)
        cap = 1 << max(4, (groups * 2 - 1).bit_length())
        ws = gq.agg_workspace(cap)
        for _ in range(2):
            gq.hash_agg(keys, vals, cap, workspace=ws, max_groups=groups + 2, ops=gq.AGG_SUM)
        gq.kernel_stats_reset()
        gq.hash_agg(keys, vals, cap, workspace=ws, max_groups=groups + 2, ops=gq.AGG_SUM)
        ms, _ = gq.kernel_stats("agg_build")
        print(f"LDS-agg groups={groups:>9,} cap={cap:>7} build={ms:7.2f}ms", flush=True)
        del keys, vals, ws
        torch.cuda.empty_cache()
