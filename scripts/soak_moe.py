"""100-step MoE-8x soak: loss falls, no memory creep, routing drift safe
(the batched-experts OOB class regression-guard at model scale)."""
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
N_STEPS = int(os.environ.get("SOAK_STEPS", "100"))
losses, mems = [], []
for it in range(N_STEPS):
    loss = m(x, labels=x)["loss"]
    loss.backward()
    opt.step(); opt.zero_grad()
    if it % 10 == 0:
        torch.cuda.synchronize()
        losses.append(loss.item())
        mems.append(torch.cuda.memory_allocated() / 1e9)
print("losses:", [round(l, 3) for l in losses])
print("mem GB:", [round(v, 2) for v in mems])
assert losses[-1] < losses[0] * 0.6
assert mems[-1] - mems[1] < 2.0   # routing drift moves buffer sizes a bit
print("SOAK MOE OK")
