"""A/B: plain two-column gather vs interleaved-pair gather at 1B rows."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from spark_amd import gpuq as gq

n = 1_000_000_000
keys = gq.gen_i64(seed=42, n=n)
pay1 = gq.gen_i64(seed=43, n=n)
pay2 = gq.gen_f64_unit(seed=44, n=n)
ws = gq.sort_workspace(n)
perm, _ = gq.sort_perm(keys, workspace=ws, out_keys=False)
out1 = torch.empty(n, dtype=torch.int64, device="cuda")
out2 = torch.empty(n, dtype=torch.float64, device="cuda")
pairs = torch.empty(n * 2, dtype=torch.int64, device="cuda")

def t(label, fn, reps=5):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps): fn()
    torch.cuda.synchronize()
    print(f"{label:28s} {(time.perf_counter()-t0)/reps*1e3:8.2f} ms", flush=True)

def plain():
    gq._check(gq.lib().gpuq_gather2_i64(gq._stream(), n, pay1.data_ptr(), pay2.data_ptr(),
                              perm.data_ptr(), out1.data_ptr(), out2.data_ptr()))
def fast():
    gq._check(gq.lib().gpuq_gather2_i64_fast(gq._stream(), n, pay1.data_ptr(), pay2.data_ptr(),
                                   perm.data_ptr(), out1.data_ptr(), out2.data_ptr(),
                                   pairs.data_ptr()))
gq.profiling(True)
t("plain gather2", plain)
t("interleaved gather2", fast)
for k in ("gather2", "interleave2", "gather2_pairs"):
    ms, cnt = gq.kernel_stats(k)
    if cnt: print(f"  {k}: {ms/cnt:.2f} ms avg x{cnt}")
# correctness
plain(); a1 = out1.clone(); a2 = out2.clone()
out1.zero_(); out2.zero_()
fast()
assert torch.equal(out1, a1) and torch.equal(out2, a2)
print("parity OK")
