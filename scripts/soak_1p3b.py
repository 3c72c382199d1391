"""100-step 1.3B soak: loss falls, no memory creep, stable step time."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from torchdistpackage_amd.models.gpt2 import GPT2Model, gpt2_xl_1p3b
from torchdistpackage_amd.ops.optim import FusedAdamW

dev = torch.device("cuda")
torch.manual_seed(0)
m = GPT2Model(gpt2_xl_1p3b(), device=dev, dtype=torch.bfloat16)
opt = FusedAdamW(m.parameters(), lr=1e-4)
x = torch.randint(0, 50304, (16, 1024), device=dev)
losses, mems, times = [], [], []
import os
N_STEPS = int(os.environ.get("SOAK_STEPS", "100"))
for it in range(N_STEPS):
    t0 = time.perf_counter()
    loss = m(x, labels=x)["loss"]
    loss.backward()
    opt.step(); opt.zero_grad()
    if it % 10 == 0:
        torch.cuda.synchronize()
        losses.append(loss.item())
        mems.append(torch.cuda.memory_allocated() / 1e9)
        times.append((time.perf_counter() - t0) * 1e3)
torch.cuda.synchronize()
print("losses:", [round(l, 3) for l in losses])
print("mem GB:", [round(v, 2) for v in mems])
assert losses[-1] < losses[0] * 0.6, "loss insufficient fall"
assert mems[-1] - mems[1] < 1.0, "memory creep"
assert all(l == l for l in losses), "nan"
print("SOAK 1.3B OK")
