import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from torchdistpackage_amd.moe import ExpertParallelMoE
dev = torch.device("cuda")
torch.manual_seed(0)
moe = ExpertParallelMoE(1024, num_experts=8, top_k=2, hidden_mult=4,
                        device=dev, dtype=torch.bfloat16)
x = torch.randn(1024, 16, 1024, dtype=torch.bfloat16, device=dev)

def t(fn, n=5):
    for _ in range(2): fn()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/n*1e3

print(f"moe layer fwd: {t(lambda: moe(x)):.2f} ms")

# pieces
xt = x.reshape(-1, 1024)
topk_idx, topk_gate, aux = moe.router(xt)
print(f"router: {t(lambda: moe.router(xt)):.2f} ms")
flat_expert = topk_idx.reshape(-1)
print(f"argsort: {t(lambda: torch.argsort(flat_expert, stable=True)):.2f} ms")
order = torch.argsort(flat_expert, stable=True)
token_of = order // 2
disp = xt[token_of]
print(f"gather disp: {t(lambda: xt[token_of]):.2f} ms")
counts = torch.bincount(flat_expert, minlength=8)
print(f"bincount+cpu: {t(lambda: torch.bincount(flat_expert, minlength=8).cpu()):.2f} ms")
seg_sizes = counts.reshape(1, -1).reshape(-1).cpu()
seg_expert = torch.arange(8) % 8
def prep():
    tok = torch.repeat_interleave(seg_expert, seg_sizes).to(dev)
    o2 = torch.argsort(tok, stable=True)
    return o2
print(f"regroup prep (cpu ri + h2d + argsort): {t(prep):.2f} ms")
o2 = prep()
g = disp[o2]
per = torch.bincount(torch.repeat_interleave(seg_expert, seg_sizes).to(dev).cpu(), minlength=8).tolist()
def experts():
    off = 0; parts=[]
    for le in range(8):
        n = per[le]
        if n: parts.append(moe.experts[le](g[off:off+n]))
        off += n
    return torch.cat(parts, 0)
print(f"experts: {t(experts):.2f} ms")
y = experts()
outs = torch.empty_like(disp)
print(f"scatter outs[o2]: {t(lambda: outs.__setitem__(o2, y)):.2f} ms")
comb = torch.zeros_like(xt)
gates = topk_gate.reshape(-1)[order].to(torch.bfloat16)
print(f"combine index_add: {t(lambda: torch.zeros_like(xt).index_add_(0, token_of, y * gates.unsqueeze(-1))):.2f} ms")
