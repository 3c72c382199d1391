"""Per-(M, shape) A/B: in-tree streaming GEMV vs hipBLASLt (F.linear)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

from torchdistpackage_amd.ops import ext

dev = torch.device("cuda")
SHAPES = [("qkv", 6144, 2048), ("proj", 2048, 2048), ("fc1", 8192, 2048),
          ("fc2", 2048, 8192), ("head", 50304, 2048)]


def t(fn, iters=50):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


e = ext("gemv")
print(f"{'shape':6s} {'M':>3s} {'mine us':>8s} {'blt us':>8s} {'mine GB/s':>9s} win")
for name, N, K in SHAPES:
    w = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
    for M in (1, 4, 8, 16, 32):
        x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        with torch.no_grad():
            mine = t(lambda: e.gemv_bf16(x, w, None))
            blt = t(lambda: F.linear(x, w))
        gbs = N * K * 2 / (mine * 1e-6) / 1e9
        print(f"{name:6s} {M:3d} {mine:8.1f} {blt:8.1f} {gbs:9.0f} "
              f"{'MINE' if mine < blt else 'blt'}")
