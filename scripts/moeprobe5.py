import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from torchdistpackage_amd.models.moe_model import MoEModel, mixtral_style_8x
dev = torch.device("cuda")
torch.manual_seed(0)
m = MoEModel(mixtral_style_8x(), device=dev, dtype=torch.bfloat16)
x = torch.randint(0, 50304, (16, 1024), device=dev)
with torch.no_grad():
    h = m.embed(x)
    for i, blk in enumerate(m.blocks[:6]):
        h = h + blk.attn(blk.ln_1(h))
        hin = blk.ln_2(h)
        moe = blk.moe
        xt = hin.reshape(-1, 1024)
        torch.cuda.synchronize()
        def T(fn):
            t0 = time.perf_counter(); r = fn(); torch.cuda.synchronize()
            return r, (time.perf_counter()-t0)*1e3
        (ri, tr) = T(lambda: moe.router(xt))
        topk_idx, topk_gate, aux = ri
        counts = torch.bincount(topk_idx.flatten(), minlength=8)
        print(f"blk {i}: router {tr:.2f}ms counts={counts.tolist()} "
              f"h absmax={h.abs().max().item():.2f} "
              f"isnan={bool(torch.isnan(h.float()).any())}")
        (_, tmoe) = T(lambda: moe(hin))
        print(f"   moe total {tmoe:.2f}ms")
        h = h + moe(hin)
