"""1.3B-scale checkpoint/resume demo: train, save, kill the process, reload
in a fresh process and verify training CONTINUES with matching losses.

    python scripts/ckpt_resume_demo.py save   /tmp/ck   # steps 0-4, save, print 5-7
    python scripts/ckpt_resume_demo.py resume /tmp/ck   # load, print steps 5-7

The save phase prints the post-checkpoint losses it observed; the resume
phase must reproduce them (small tolerance: the attention backward uses
atomics, so bf16 training is not bitwise deterministic across runs).
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from torchdistpackage_amd import fix_rand
from torchdistpackage_amd.models.gpt2 import GPT2Config, GPT2Model
from torchdistpackage_amd.ops.optim import FusedAdamW
from torchdistpackage_amd.ddp import Bf16ZeroOptimizer
from torchdistpackage_amd.dist.checkpoint import save_checkpoint, load_checkpoint

CFG = GPT2Config(vocab_size=50304, n_layer=24, n_head=32, dim=2048,
                 max_seq=1024)


def batch(step, dev):
    g = torch.Generator(device="cpu").manual_seed(9000 + step)
    return torch.randint(0, CFG.vocab_size, (4, 1024), generator=g).to(dev)


def run_steps(model, opt, dev, steps):
    out = []
    for s in steps:
        loss = model(batch(s, dev), labels=batch(s, dev))["loss"]
        loss.backward()
        opt.step()
        opt.zero_grad()
        out.append(round(loss.item(), 4))
    return out


def main():
    phase, ckdir = sys.argv[1], sys.argv[2]
    dev = torch.device("cuda")
    fix_rand(0)
    model = GPT2Model(CFG, device=dev, dtype=torch.bfloat16)
    opt = Bf16ZeroOptimizer(FusedAdamW(model.parameters(), lr=1e-4))
    if phase == "save":
        run_steps(model, opt, dev, range(5))
        save_checkpoint(ckdir, 5, model, optimizer=opt)
        print("post-ckpt losses:", run_steps(model, opt, dev, range(5, 8)))
    else:
        load_checkpoint(ckdir, model, optimizer=opt)
        print("post-ckpt losses:", run_steps(model, opt, dev, range(5, 8)))
    print(f"CKPT {phase.upper()} OK")


if __name__ == "__main__":
    main()
