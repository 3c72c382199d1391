"""50-step training sanity: loss must trend down on a repeating synthetic
batch (overfit check — validates the whole fwd+bwd+optimizer loop)."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from torchdistpackage_amd.models.gpt2 import GPT2Config, GPT2Model
from torchdistpackage_amd.ops.optim import FusedAdamW

dev = torch.device("cuda")
cfg = GPT2Config(vocab_size=2048, n_layer=6, n_head=8, dim=512, max_seq=256)
torch.manual_seed(0)
m = GPT2Model(cfg, device=dev, dtype=torch.bfloat16)
opt = FusedAdamW(m.parameters(), lr=3e-4)
x = torch.randint(0, 2048, (4, 256), device=dev)
losses = []
for it in range(50):
    loss = m(x, labels=x)["loss"]
    loss.backward()
    opt.step(); opt.zero_grad()
    losses.append(loss.item())
print(f"loss[0]={losses[0]:.3f} loss[10]={losses[10]:.3f} "
      f"loss[49]={losses[49]:.3f}")
assert losses[49] < losses[0] * 0.5, "loss did not fall enough"
assert all(l == l for l in losses), "nan loss"
print("soak OK")
