"""Scatter phase ablation: which phase owns the parked time?
MODE 0 loads | 1 +rank | 2 +stage | 3 +drain (full shape)."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import ctypes
import torch
from spark_amd import gpuq as gq

n = 500_000_000
keys = gq.gen_i64(seed=1, n=n)           # raw bits as u64 keys
idx = torch.arange(n, dtype=torch.int32, device="cuda")
kout = torch.empty(n, dtype=torch.int64, device="cuda")
iout = torch.empty(n, dtype=torch.int32, device="cuda")
gbase = (torch.arange(256, dtype=torch.int64, device="cuda") * (n // 256)).to(torch.int32)
sink = torch.zeros(1, dtype=torch.int64, device="cuda")
L = gq.lib()
L.gpuq_scatter_ablate.restype = ctypes.c_int32
L.gpuq_scatter_ablate.argtypes = [ctypes.c_void_p, ctypes.c_int64] + [ctypes.c_void_p] * 6 + [ctypes.c_int32]

for mode, name in [(0, "loads"), (1, "+rank"), (2, "+stage"), (3, "+drain(full)")]:
    def run():
        rc = L.gpuq_scatter_ablate(gq._stream(), n, keys.data_ptr(), idx.data_ptr(),
                                   kout.data_ptr(), iout.data_ptr(),
                                   gbase.data_ptr(), sink.data_ptr(), mode)
        assert rc == 0
    run(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(5):
        run()
    torch.cuda.synchronize()
    print(f"mode {mode} {name:14s} {(time.perf_counter()-t0)/5*1e3:7.2f} ms", flush=True)

# drain store shape A/B (512x8 tile 4096): two stores (12B) vs one 16B record
rout = torch.empty(n * 2, dtype=torch.int64, device="cuda")  # 16B records
L.gpuq_scatter_ablate2.restype = ctypes.c_int32
L.gpuq_scatter_ablate2.argtypes = [ctypes.c_void_p, ctypes.c_int64] + [ctypes.c_void_p] * 6 + [ctypes.c_int32]
for mode, name in [(4, "two-stores-12B"), (5, "one-16B-record")]:
    def run2():
        rc = L.gpuq_scatter_ablate2(gq._stream(), n, keys.data_ptr(), idx.data_ptr(),
                                    kout.data_ptr(), iout.data_ptr(), rout.data_ptr(),
                                    gbase.data_ptr(), mode)
        assert rc == 0
    run2(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(5):
        run2()
    torch.cuda.synchronize()
    print(f"mode {mode} {name:15s} {(time.perf_counter()-t0)/5*1e3:7.2f} ms", flush=True)
