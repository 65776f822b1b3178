"""Where does sort wall time go? Phase-by-phase with explicit syncs."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from spark_amd import gpuq as gq

n = 1_000_000_000
keys = gq.gen_i64(seed=42, n=n)
pay1 = gq.gen_i64(seed=43, n=n)
pay2 = gq.gen_f64_unit(seed=44, n=n)
ws = gq.sort_workspace(n)
out1 = torch.empty(n, dtype=torch.int64, device="cuda")
out2 = torch.empty(n, dtype=torch.float64, device="cuda")

def t(label, fn, reps=3):
    fn()  # warm
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    print(f"{label:30s} {(time.perf_counter()-t0)/reps*1e3:8.2f} ms", flush=True)

state = {}
def full_sort():
    state['perm'], state['skeys'] = gq.sort_perm(keys, workspace=ws)
t("sort_perm (incl out_keys)", full_sort)
def sort_nokeys():
    state['perm'], _ = gq.sort_perm(keys, workspace=ws, out_keys=False)
t("sort_perm (perm only)", sort_nokeys)
perm = state['perm']
def gathers():
    gq.lib().gpuq_gather(gq._stream(), n, gq._col(pay1), perm.data_ptr(), out1.data_ptr())
    gq.lib().gpuq_gather(gq._stream(), n, gq._col(pay2), perm.data_ptr(), out2.data_ptr())
t("2x payload gather", gathers)
def step():
    full_sort(); gathers()
t("full step", step)
