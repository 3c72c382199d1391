import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from torchdistpackage_amd.ops import rope_rotate_half, swiglu, batched_bias_gelu, ext
dev="cuda"
B,H,S,D = 8,32,1024,128
inv = 1.0/(500000.0 ** (torch.arange(0,D,2).float()/D))
fr = torch.outer(torch.arange(S).float(), inv)
cos, sin = fr.cos().cuda(), fr.sin().cuda()
x = torch.randn(B,H,S,D,device=dev,dtype=torch.bfloat16)
a = torch.randn(8192,14336,device=dev,dtype=torch.bfloat16); b = torch.randn_like(a)
h3 = torch.randn(8, 4608, 3072, device=dev, dtype=torch.bfloat16)
bb = torch.randn(8, 3072, device=dev, dtype=torch.bfloat16)
ema = torch.zeros(100_000_000, device=dev); p = torch.zeros(100_000_000, device=dev, dtype=torch.bfloat16)
for _ in range(5):
    rope_rotate_half(x, cos, sin)
    swiglu(a, b)
    batched_bias_gelu(h3, bb)
    ext("e").ema_update(ema, p, 0.999)
torch.cuda.synchronize(); print("done")
