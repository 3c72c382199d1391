import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from torchdistpackage_amd.models.moe_model import MoEModel, mixtral_style_8x
from torchdistpackage_amd.ops.optim import FusedAdamW
dev = torch.device("cuda")
torch.manual_seed(0)
m = MoEModel(mixtral_style_8x(), device=dev, dtype=torch.bfloat16)
opt = FusedAdamW(m.parameters(), lr=1e-4)
x = torch.randint(0, 50304, (16, 1024), device=dev)

def timeit(fn, n=3):
    fn(); torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/n*1e3

def fwd(): m(x, labels=x)
def fwdbwd():
    out = m(x, labels=x); out["loss"].backward(); m.zero_grad(set_to_none=True)
def full():
    out = m(x, labels=x); out["loss"].backward(); opt.step(); opt.zero_grad()
print(f"fwd {timeit(fwd):.1f}ms  fwdbwd {timeit(fwdbwd):.1f}ms  full {timeit(full):.1f}ms")
