"""FSDP2 fully_shard + CPU-offload memory experiment.

Reference parity: /root/reference/examples/fsdp2_offload_test.py:32-114 —
torch FSDP2 is used as-is (example tier, nothing in-package), with the
before/after memory report.

Launch: torchrun --nproc-per-node N --master-addr 127.0.0.1 examples/fsdp2_offload.py
"""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from torchdistpackage_amd import setup_distributed, fix_rand
from torchdistpackage_amd.models.gpt2 import GPT2Model, GPT2Config


def report(tag):
    if torch.cuda.is_available():
        print(f"[{tag}] allocated "
              f"{torch.cuda.memory_allocated() / 1e9:.2f} GB, reserved "
              f"{torch.cuda.memory_reserved() / 1e9:.2f} GB")


def main():
    info = setup_distributed()
    fix_rand(info["rank"])
    if not torch.cuda.is_available():
        print("fsdp2 offload example needs GPUs")
        return
    from torch.distributed.fsdp import fully_shard, CPUOffloadPolicy

    cfg = GPT2Config(n_layer=4, n_head=8, dim=512, max_seq=256)
    model = GPT2Model(cfg, device="cuda", dtype=torch.bfloat16)
    report("before shard")
    for blk in model.blocks:
        fully_shard(blk, offload_policy=CPUOffloadPolicy())
    fully_shard(model, offload_policy=CPUOffloadPolicy())
    report("after shard+offload")
    x = torch.randint(0, cfg.vocab_size, (2, 256), device="cuda")
    loss = model(x, labels=x)["loss"]
    loss.backward()
    report("after fwd+bwd")
    print("loss", loss.item())


if __name__ == "__main__":
    main()
    import torch.distributed as _d
    if _d.is_initialized():
        _d.destroy_process_group()
