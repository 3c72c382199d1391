import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from torchdistpackage_amd.models.moe_model import MoEModel, mixtral_style_8x
dev = torch.device("cuda")
torch.manual_seed(0)
m = MoEModel(mixtral_style_8x(), device=dev, dtype=torch.bfloat16)
x = torch.randint(0, 50304, (16, 1024), device=dev)
with torch.no_grad():
    m(x)  # warm
torch.cuda.synchronize()
st0 = torch.cuda.memory_stats()
h = m.embed(x)
torch.cuda.synchronize()
times = []
with torch.no_grad():
    for i, blk in enumerate(m.blocks):
        t0 = time.perf_counter()
        # time attn and moe inside separately
        a0 = time.perf_counter()
        hh = blk.ln_1(h); hh = blk.attn(hh)
        torch.cuda.synchronize(); a1 = time.perf_counter()
        h = h + hh
        g0 = time.perf_counter()
        mm = blk.ln_2(h); mm = blk.moe(mm)
        torch.cuda.synchronize(); g1 = time.perf_counter()
        h = h + mm
        times.append((a1-a0, g1-g0))
print("first 6 blocks (attn_ms, moe_ms):",
      [(round(a*1e3,2), round(g*1e3,2)) for a, g in times[:6]])
print("sum:", round(sum(a+g for a,g in times)*1e3, 1), "ms")
st1 = torch.cuda.memory_stats()
print("device mallocs during loop:",
      st1.get('num_device_alloc',0) - st0.get('num_device_alloc',0),
      "frees:", st1.get('num_device_free',0) - st0.get('num_device_free',0))
