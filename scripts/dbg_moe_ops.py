import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from torchdistpackage_amd.ops import bias_gelu
dev = "cuda"
def ck(tag):
    torch.cuda.synchronize(); print(tag, "ok", flush=True)

torch.manual_seed(0)
cnt = torch.randint(0, 8192, (8,), device=dev)
total = int(cnt.sum())
print("cnt", cnt.tolist(), "total", total, flush=True)
g = (torch.randn(total, 2048, device=dev) * 0.1).bfloat16()
ck("alloc")
maxn = int(cnt.max()); Q=512; maxn_pad = max((maxn+Q-1)//Q*Q, Q)
offs = torch.cumsum(cnt,0)-cnt
ar = torch.arange(maxn_pad, device=dev)
idx = (offs[:,None] + torch.minimum(ar[None,:], (cnt[:,None]-1).clamp(min=0))).clamp_(0, total-1)
xg = g.index_select(0, idx.reshape(-1)).view(8, maxn_pad, 2048)
ck("gather")
w1 = (torch.randn(8, 8192, 2048, device=dev)*0.02).bfloat16()
b1 = torch.zeros(8, 8192, device=dev).bfloat16()
h = torch.baddbmm(b1.unsqueeze(1), xg, w1.transpose(1,2))
ck("baddbmm1")
hg = bias_gelu(h, None)
ck("bias_gelu")
w2 = (torch.randn(8, 2048, 8192, device=dev)*0.02).bfloat16()
b2 = torch.zeros(8, 2048, device=dev).bfloat16()
y = torch.baddbmm(b2.unsqueeze(1), hg, w2.transpose(1,2))
ck("baddbmm2")
valid = ar[None,:] < cnt[:,None]
ya = y.reshape(-1, 2048)[valid.reshape(-1)]
ck("mask")
# backward pieces
g2 = g.clone().requires_grad_(True)
xg2 = g2.index_select(0, idx.reshape(-1)).view(8, maxn_pad, 2048)
h2 = torch.baddbmm(b1.unsqueeze(1).float().bfloat16(), xg2, w1.transpose(1,2))
hg2 = bias_gelu(h2, None)
y2 = torch.baddbmm(b2.unsqueeze(1), hg2, w2.transpose(1,2))
ya2 = y2.reshape(-1,2048)[valid.reshape(-1)]
ya2.sum().backward()
ck("backward")
print("DONE", flush=True)
