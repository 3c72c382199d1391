import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))) if '__file__' in dir() else '.')
sys.path.insert(0, '.')
import torch
from torchdistpackage_amd.ops import ext
e = ext("gemm")
def bench_one(fn, iters=15, warmup=5):
    for _ in range(warmup): fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True); t = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters): fn()
    t.record(); torch.cuda.synchronize()
    return s.elapsed_time(t)/iters*1e-3
torch.manual_seed(0)
for N in (4096, 8192):
    x = (torch.rand(N, N, device="cuda", dtype=torch.float32)*2-1).bfloat16()
    w = (torch.rand(N, N, device="cuda", dtype=torch.float32)*2-1).bfloat16()
    fl = 2.0*N*N*N
    tm = bench_one(lambda: e.gemm_fprop(x, w, None))
    tl = bench_one(lambda: x @ w.t())
    print(f"{N}^3 uniform[-1,1): mine={fl/tm/1e12:.0f}TF hipblaslt={fl/tl/1e12:.0f}TF")
