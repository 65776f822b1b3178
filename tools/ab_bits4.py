"""A/B: 8-bit onesweep vs 4-bit cooperative-lookback mode at 1B rows."""
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
    print(f"{label:24s} {ms:8.2f} ms/sort", flush=True)
    return ms

gq.profiling(True)
t("8-bit onesweep")
sms, scnt = gq.kernel_stats("radix_scatter")
print(f"  scatter {sms/max(scnt,1):.3f} ms/pass x {scnt}")
gq.kernel_stats_reset()
os.environ["GPUQ_SORT_BITS"] = "4"
t("4-bit coop lookback")
sms, scnt = gq.kernel_stats("radix_scatter")
print(f"  scatter {sms/max(scnt,1):.3f} ms/pass x {scnt}")

# parity: bit-exact same permutation
m = 100_000_000
kk = gq.gen_i64(seed=7, n=m)
ws2 = gq.sort_workspace(m)
p4, _ = gq.sort_perm(kk, workspace=ws2, out_keys=False)
del os.environ["GPUQ_SORT_BITS"]
p8, _ = gq.sort_perm(kk, workspace=ws2, out_keys=False)
assert torch.equal(p4, p8), "4-bit vs 8-bit permutation mismatch"
print("parity OK (100M rows, bit-exact permutation)")
