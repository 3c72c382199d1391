"""GEMM A/B at the 8-GPU flagship's PER-RANK shapes (dp2·pp2·tp2, B_mb=1,
S=1024): decides the dispatch policy the driver's SCALE run will hit."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from torchdistpackage_amd.ops import ext
from torchdistpackage_amd.ops.gemm import pick_splitk
e = ext("gemm")
def t(fn, iters=30, warmup=8):
    for _ in range(warmup): fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True); q = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters): fn()
    q.record(); torch.cuda.synchronize()
    return s.elapsed_time(q)/iters*1e-3
torch.manual_seed(0)
M = 1024  # tokens per micro-batch per rank
shapes = [("qkv-f","f",M,3072,2048),("proj-f","f",M,2048,1024),
          ("fc1-f","f",M,4096,2048),("fc2-f","f",M,2048,4096),
          ("qkv-w","w",3072,2048,M),("proj-w","w",2048,1024,M),
          ("fc1-w","w",4096,2048,M),("fc2-w","w",2048,4096,M),
          ("qkv-d","d",M,2048,3072),("fc2-d","d",M,4096,2048)]
for tag,kind,Mm,N,K in shapes:
    fl = 2.0*Mm*N*K
    if kind=="f":
        x = torch.randn(Mm,K,device="cuda",dtype=torch.bfloat16)
        w = torch.randn(N,K,device="cuda",dtype=torch.bfloat16)
        mine = lambda: e.gemm_fprop(x,w,None); lib = lambda: x@w.t()
        sk = 1
    elif kind=="w":
        dy = torch.randn(K,Mm,device="cuda",dtype=torch.bfloat16)
        xx = torch.randn(K,N,device="cuda",dtype=torch.bfloat16)
        sk = pick_splitk(Mm,N,K)
        mine = lambda: e.gemm_wgrad(dy,xx,sk,True); lib = lambda: dy.t()@xx
    else:
        dy = torch.randn(Mm,K,device="cuda",dtype=torch.bfloat16)
        w = torch.randn(K,N,device="cuda",dtype=torch.bfloat16)
        mine = lambda: e.gemm_dgrad(dy,w,True); lib = lambda: dy@w
        sk = 1
    tm, tl = t(mine), t(lib)
    print(f"{tag:7s} {Mm}x{N}x{K} sk{sk}: mine={fl/tm/1e12:6.0f}TF ({tm*1e6:5.0f}us) lib={fl/tl/1e12:6.0f}TF ({tl*1e6:5.0f}us) ratio={tl/tm:5.2f}")
