import sys, time, torch, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torchdistpackage_amd.ops as ops
import torch.nn.functional as F
dev = torch.device("cuda")

def t(fn, n=20):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/n*1e6

R, DIN, DOUT = 16384, 2048, 6144
x = torch.randn(R, DIN, dtype=torch.bfloat16, device=dev, requires_grad=True)
w = torch.randn(DOUT, DIN, dtype=torch.bfloat16, device=dev, requires_grad=True)
b = torch.randn(DOUT, dtype=torch.bfloat16, device=dev, requires_grad=True)
dy = torch.randn(R, DOUT, dtype=torch.bfloat16, device=dev)

def ref_step():
    y = F.linear(x, w, b)
    torch.autograd.grad(y, [x, w, b], dy)
def lb_step():
    y = ops.linear_bias(x, w, b)
    torch.autograd.grad(y, [x, w, b], dy)
print(f"F.linear fwd+bwd us: {t(ref_step):.1f}")
print(f"linear_bias fwd+bwd us: {t(lb_step):.1f}")

dy2 = dy; x2 = x.detach()
print(f"wgrad dy.t@x us: {t(lambda: dy2.t() @ x2):.1f}")
print(f"wgrad via matmul(dy.mT, x) us: {t(lambda: torch.matmul(dy2.mT, x2)):.1f}")
o = torch.empty(DOUT, DIN, dtype=torch.bfloat16, device=dev)
print(f"wgrad mm out= us: {t(lambda: torch.mm(dy2.t(), x2, out=o)):.1f}")
print(f"dgrad dy@w us: {t(lambda: dy2 @ w.detach()):.1f}")
print(f"colsum dy us: {t(lambda: ops.colsum(dy2)):.1f}")
print(f"torch dbias sum us: {t(lambda: dy2.float().sum(0)):.1f}")
print(f"torch dbias sum bf16 us: {t(lambda: dy2.sum(0)):.1f}")
