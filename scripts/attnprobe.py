import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from torchdistpackage_amd.parallel.tensor import Attention
from torchdistpackage_amd.ops import fused_qkv_attention, flash_attention
dev = torch.device("cuda")
torch.manual_seed(0)

def t(fn, n=5):
    for _ in range(2): fn()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/n*1e3

for (dim, nh) in ((1024, 8), (2048, 16)):
    a = Attention(dim, nh, causal=True, device=dev, dtype=torch.bfloat16)
    x = torch.randn(1024, 16, dim, dtype=torch.bfloat16, device=dev)
    print(f"dim={dim} nh={nh} attn module: {t(lambda: a(x)):.2f} ms")
    qkv = torch.randn(1024, 16, 3*dim, dtype=torch.bfloat16, device=dev)
    print(f"  fused_qkv_attention: {t(lambda: fused_qkv_attention(qkv, nh)):.2f} ms")
    q = torch.randn(16, nh, 1024, dim//nh, dtype=torch.bfloat16, device=dev)
    print(f"  flash (contig {dim//nh}): {t(lambda: flash_attention(q,q,q)):.2f} ms")
