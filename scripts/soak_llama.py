"""100-step Llama-8B + ZeRO + ShardedEMA soak: loss falls, no memory creep
(exercises fused RoPE/SwiGLU/EMA, zero-copy attention, batched ZeRO copies)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from torchdistpackage_amd.models.llama import LlamaModel, llama3_8b
from torchdistpackage_amd.ops.optim import FusedAdamW
from torchdistpackage_amd.ddp import Bf16ZeroOptimizer
from torchdistpackage_amd.dist.sharded_ema import ShardedEMA

dev = torch.device("cuda")
torch.manual_seed(0)
m = LlamaModel(llama3_8b(), device=dev, dtype=torch.bfloat16)
inner = FusedAdamW(m.parameters(), lr=1e-4)
opt = Bf16ZeroOptimizer(inner, stage2=True)
ema = ShardedEMA(m, decay=0.999)
x = torch.randint(0, 128256, (4, 1024), device=dev)
N_STEPS = int(os.environ.get("SOAK_STEPS", "100"))
losses, mems = [], []
for it in range(N_STEPS):
    loss = m(x, labels=x)["loss"]
    loss.backward()
    opt.step(); opt.zero_grad(); ema.update()
    if it % 10 == 0:
        torch.cuda.synchronize()
        losses.append(loss.item())
        mems.append(torch.cuda.memory_allocated() / 1e9)
print("losses:", [round(l, 3) for l in losses])
print("mem GB:", [round(v, 2) for v in mems])
assert losses[-1] < losses[0] * 0.6
assert mems[-1] - mems[1] < 1.0
assert all(l == l for l in losses)
print("SOAK LLAMA OK")
