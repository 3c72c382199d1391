import sys, time, torch
sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
import torchdistpackage_amd.ops as ops
dev = torch.device("cuda")

def t(fn, n=20):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/n*1e6

# LN bwd at bench shape (B=16)
x = torch.randn(16384, 2048, dtype=torch.bfloat16, device=dev, requires_grad=True)
w = torch.randn(2048, dtype=torch.bfloat16, device=dev, requires_grad=True)
b = torch.randn(2048, dtype=torch.bfloat16, device=dev, requires_grad=True)
y = ops.layer_norm(x, w, b); dy = torch.randn_like(y)
g = torch.autograd.grad(y, [x,w,b], dy, retain_graph=True)
print(f"ln fwd+bwd pair us: fwd={t(lambda: ops.ext('x').layernorm_fwd(x, w, b, 1e-5)):.1f}")
mean  = torch.randn(16384, device=dev); rstd = torch.rand(16384, device=dev)+0.5
print(f"ln bwd us: {t(lambda: ops.ext('x').layernorm_bwd(dy, x, w, mean, rstd)):.1f}")

# attention at bench shape (B=16, H=16, S=1024, D=128)
q = torch.randn(16,16,1024,128, dtype=torch.bfloat16, device=dev)
k = torch.randn_like(q); v = torch.randn_like(q); o = torch.empty_like(q)
print(f"attn fwd us: {t(lambda: ops.ext('x').attn_fwd(q,k,v,o,True,0.088)):.1f}")
_, lse = ops.ext('x').attn_fwd(q,k,v,o,True,0.088)
do = torch.randn_like(q); dq=torch.empty_like(q); dk=torch.empty_like(q); dv=torch.empty_like(q)
print(f"attn bwd us: {t(lambda: ops.ext('x').attn_bwd(do,q,k,v,o,lse,dq,dk,dv,True,0.088)):.1f}")

# bias_gelu at bench fc1 shape (16384, 8192)
xg = torch.randn(16384, 8192, dtype=torch.bfloat16, device=dev)
bg = torch.randn(8192, dtype=torch.bfloat16, device=dev)
dyg = torch.randn_like(xg)
print(f"bias_gelu fwd us: {t(lambda: ops.ext('x').bias_gelu_fwd(xg, bg)):.1f}")
print(f"bias_gelu bwd us: {t(lambda: ops.ext('x').bias_gelu_bwd(dyg, xg, bg)):.1f}")

# fused AdamW at 1.3B-param scale (bf16 params + fp32 master, bf16 grads)
from torchdistpackage_amd.ops.optim import FusedAdamW
pa = torch.nn.Parameter(torch.zeros(1_300_000_000, dtype=torch.bfloat16,
                                    device=dev))
pa.grad = torch.randn_like(pa.data)
oa = FusedAdamW([pa], lr=1e-3)
oa.step()
print(f"adamw 1.3e9 us: {t(lambda: oa.step(), n=10):.1f}")
