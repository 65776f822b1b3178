"""A/B: strided drain vs contiguous paired-store drain (sort scatter)."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from spark_amd import gpuq as gq

n = 1_000_000_000
keys = gq.gen_i64(seed=42, n=n)
ws = gq.sort_workspace(n)

def t(label, reps=5):
    gq.kernel_stats_reset()
    gq.sort_perm(keys, workspace=ws, out_keys=False)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        gq.sort_perm(keys, workspace=ws, out_keys=False)
    torch.cuda.synchronize()
    ms = (time.perf_counter()-t0)/reps*1e3
    sms, scnt = gq.kernel_stats("radix_scatter")
    print(f"{label:30s} {ms:8.2f} ms/sort  scatter {sms/max(scnt,1):.3f} ms/pass", flush=True)

gq.profiling(True)
t(f"geom={os.environ.get('GPUQ_SORT_GEOM','512x10')} contig={bool(os.environ.get('GPUQ_DRAIN_CONTIG'))}")
# parity vs reference run handled by the existing suite; quick check here:
# verify the result IS a sorted permutation (a silent geometry fallback
# once produced fast-but-wrong timings — never trust a sweep without this)
perm, skeys = gq.sort_perm(keys, workspace=ws)
assert bool((skeys[1:] >= skeys[:-1]).all()), "output not sorted!"
idx = torch.randint(0, n, (1_000_000,), device="cuda")
assert bool((skeys[idx] == keys[perm[idx].long()]).all()), "not a permutation!"
print("sorted-permutation check OK")
