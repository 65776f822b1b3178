"""Small probe for PMC counter runs. ONLY idempotent kernels: rocprofv3
counter collection replays dispatches, which deadlocks the decoupled-
lookback sort (consumed epoch state) — so the permutation comes from
torch.randint (tools-only; the product path never computes in torch)."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from spark_amd import gpuq as gq

n = 100_000_000
pay1 = gq.gen_i64(seed=43, n=n)
pay2 = gq.gen_f64_unit(seed=44, n=n)
perm = torch.randint(0, n, (n,), dtype=torch.int32, device="cuda")
out1 = torch.empty(n, dtype=torch.int64, device="cuda")
out2 = torch.empty(n, dtype=torch.float64, device="cuda")
pairs = torch.empty(n * 2, dtype=torch.int64, device="cuda")
gq._check(gq.lib().gpuq_gather2_i64_fast(gq._stream(), n, pay1.data_ptr(),
          pay2.data_ptr(), perm.data_ptr(), out1.data_ptr(), out2.data_ptr(),
          pairs.data_ptr()))
gq._check(gq.lib().gpuq_gather2_i64(gq._stream(), n, pay1.data_ptr(),
          pay2.data_ptr(), perm.data_ptr(), out1.data_ptr(), out2.data_ptr()))
torch.cuda.synchronize()
print("pmc probe done")
