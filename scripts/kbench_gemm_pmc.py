"""Single-shape GEMM loops for rocprofv3 PMC collection (no timing)."""
import os
import sys
import torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from torchdistpackage_amd.ops import ext

kind = sys.argv[1] if len(sys.argv) > 1 else "fprop"
e = ext("gemm")
torch.manual_seed(0)
T = 16384
if kind == "fprop":   # qkv fprop
    x = torch.randn(T, 2048, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(6144, 2048, device="cuda", dtype=torch.bfloat16)
    fn = lambda: e.gemm_fprop(x, w, None)
elif kind == "wgrad":  # fc1 wgrad (sk=1, 256 tiles)
    dy = torch.randn(T, 8192, device="cuda", dtype=torch.bfloat16)
    xx = torch.randn(T, 2048, device="cuda", dtype=torch.bfloat16)
    fn = lambda: e.gemm_wgrad(dy, xx, 1, True)
for _ in range(10):
    fn()
torch.cuda.synchronize()
print("done", kind)
