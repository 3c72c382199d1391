"""Microbench of the round-2 kernels at their bench shapes."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from torchdistpackage_amd.ops import (rope_rotate_half, swiglu, ext)

def t(fn, n=30):
    for _ in range(8): fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True); q = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(n): fn()
    q.record(); torch.cuda.synchronize()
    return s.elapsed_time(q)/n*1e3  # us

dev="cuda"
# rope at Llama-8B q shape (B8 H32 S1024 D128)
B,H,S,D = 8,32,1024,128
inv = 1.0/(500000.0 ** (torch.arange(0,D,2).float()/D))
fr = torch.outer(torch.arange(S).float(), inv)
cos, sin = fr.cos().cuda(), fr.sin().cuda()
x = torch.randn(B,H,S,D,device=dev,dtype=torch.bfloat16)
us = t(lambda: rope_rotate_half(x, cos, sin))
gb = B*H*S*D*2*2/1e9 + B*H*S*D//2*4*2/1e9
print(f"rope fwd {B}x{H}x{S}x{D}: {us:.1f} us = {gb/us*1e6:.2f} TB/s")
# swiglu at fc shape (8192, 14336)
a = torch.randn(8192,14336,device=dev,dtype=torch.bfloat16)
b = torch.randn_like(a)
us = t(lambda: swiglu(a,b))
gb = a.numel()*2*3/1e9
print(f"swiglu fwd 8192x14336: {us:.1f} us = {gb/us*1e6:.2f} TB/s")
# ce_partial at tp2 flagship shape (16384, 25152)
lg = torch.randn(16384, 25152, device=dev, dtype=torch.bfloat16)
tg = torch.randint(0, 25152, (16384,), device=dev)
e = ext("ce")
us = t(lambda: e.ce_partial_fwd(lg, tg))
gb = lg.numel()*2/1e9
print(f"ce_partial fwd 16384x25152: {us:.1f} us = {gb/us*1e6:.2f} TB/s")
# ema bf16 at 8B shard
ema = torch.zeros(2_000_000_000, device=dev)
p = torch.zeros(2_000_000_000, device=dev, dtype=torch.bfloat16)
us = t(lambda: e.ema_update(ema, p, 0.999), n=10)
gb = (2e9*4*2 + 2e9*2)/1e9
print(f"ema bf16 2e9: {us:.1f} us = {gb/us*1e6:.2f} TB/s")
