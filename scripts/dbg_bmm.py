import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
dev="cuda"
def ck(t):
    torch.cuda.synchronize(); print(t, "ok", flush=True)
for (E,N,K,H) in [(8,512,2048,8192),(8,2048,2048,8192),(8,8192,2048,8192)]:
    x = (torch.randn(E,N,K,device=dev)*0.1).bfloat16(); ck(f"alloc x {N}")
    w = (torch.randn(E,H,K,device=dev)*0.02).bfloat16(); ck(f"alloc w {N}")
    y1 = torch.bmm(x, w.transpose(1,2)); ck(f"bmm-strided {N}")
    wc = w.transpose(1,2).contiguous(); ck(f"contig {N}")
    y2 = torch.bmm(x, wc); ck(f"bmm-contig {N}")
    print((y1.float()-y2.float()).abs().max().item(), flush=True)
print("DONE", flush=True)
