import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from torchdistpackage_amd.moe.layer import BatchedExperts, ExpertParallelMoE

dev = "cuda"
print("-- BatchedExperts unit at bench scale --")
be = BatchedExperts(8, 2048, 4, device=dev, dtype=torch.bfloat16)
for trial in range(8):
    torch.manual_seed(trial)
    cnt = torch.randint(0, 8192, (8,), device=dev)
    if trial == 3: cnt[5:] = 0
    if trial == 4: cnt[:] = 0; cnt[0] = 32768
    total = int(cnt.sum())
    g = (torch.randn(total, 2048, device=dev) * 0.1).bfloat16().requires_grad_(True)
    y = be(g, cnt, int(cnt.max()))
    y.sum().backward()
    torch.cuda.synchronize()
    print("unit trial", trial, "ok", total)

print("-- full layer at bench scale --")
torch.manual_seed(0)
moe = ExpertParallelMoE(2048, 8, 2, 4, ep_group=None, batched=True,
                        device=dev, dtype=torch.bfloat16)
for step in range(4):
    x = (torch.randn(16384, 2048, device=dev) * 0.1).bfloat16().requires_grad_(True)
    y = moe(x)
    (y.float().pow(2).mean() + moe.aux_loss).backward()
    torch.cuda.synchronize()
    print("layer step", step, "ok")

print("-- MoEModel 3 steps --")
from torchdistpackage_amd.models.moe_model import MoEModel, moe_8x
from torchdistpackage_amd.ops.optim import FusedAdamW
torch.manual_seed(0)
m = MoEModel(moe_8x(), device=dev, dtype=torch.bfloat16)
opt = FusedAdamW(m.parameters(), lr=1e-4)
xx = torch.randint(0, 50304, (16, 1024), device=dev)
for step in range(3):
    loss = m(xx, labels=xx)["loss"]
    loss.backward()
    opt.step(); opt.zero_grad()
    torch.cuda.synchronize()
    print("model step", step, "ok", float(loss))
print("ALL OK")
