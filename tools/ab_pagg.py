"""A/B: direct atomic agg vs single-pass bucket-partitioned agg, config 3."""
import sys, os, time, ctypes
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from spark_amd import gpuq as gq

n, groups = 1_000_000_000, int(os.environ.get("AB_GROUPS", 10_000_000))
cap = 1 << (groups * 2 - 1).bit_length()
keys = gq.gen_i64(seed=52, n=n, range_=groups)
vals = gq.gen_f64_unit(seed=53, n=n)
ws = gq.agg_workspace(cap)
pws = torch.empty(gq.lib().gpuq_hash_agg_part_workspace_bytes(n, cap),
                  dtype=torch.uint8, device="cuda")
mg = groups + 2
outs = [torch.empty(mg, dtype=d, device="cuda")
        for d in (torch.int64, torch.uint8, torch.float64, torch.uint8, torch.int64)]

def t(label, fn, reps=5):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps): fn()
    torch.cuda.synchronize()
    print(f"{label:20s} {(time.perf_counter()-t0)/reps*1e3:8.2f} ms", flush=True)

def direct():
    ng = ctypes.c_int64(0)
    gq._check(gq.lib().gpuq_hash_agg_i64_f64(
        gq._stream(), n, gq._col(keys), gq._col(vals), ws.data_ptr(), cap,
        1, 1, 1, *[x.data_ptr() for x in outs], ctypes.byref(ng)))
    return ng.value

def part():
    ng = ctypes.c_int64(0)
    gq._check(gq.lib().gpuq_hash_agg_partitioned(
        gq._stream(), n, gq._col(keys), gq._col(vals), pws.data_ptr(), cap,
        1, *[x.data_ptr() for x in outs], ctypes.byref(ng)))
    return ng.value

gq.profiling(True)
t("direct", direct)
os.environ["GPUQ_PAGG"] = os.environ.get("AB_PAGG", "1")
t("partitioned v" + os.environ["GPUQ_PAGG"], part)
for k in ("pagg_ghist", "pagg_scatter", "pagg_chunks", "agg_build"):
    ms, cnt = gq.kernel_stats(k)
    if cnt: print(f"  {k}: {ms/cnt:.2f} ms avg x{cnt}")
# parity: compare sums
g1 = direct(); k1 = outs[0][:g1].clone(); s1 = outs[2][:g1].clone()
g2 = part();   k2 = outs[0][:g2].clone(); s2 = outs[2][:g2].clone()
assert g1 == g2, (g1, g2)
o1 = torch.argsort(k1); o2 = torch.argsort(k2)
assert torch.equal(k1[o1], k2[o2])
assert torch.allclose(s1[o1], s2[o2], rtol=1e-9)
print(f"parity OK ({g1} groups)")

# variant sweep via GPUQ_PAGG (static-cached per process: report only)
