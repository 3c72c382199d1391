"""Pure-DP training with NaiveDdp vs TorchDDP step-time comparison.

Reference parity: /root/reference/examples/test_ddp.py (the differential
method lives in tests/test_naive_ddp.py; this example is the runnable
composition + timing comparison on real GPUs).

Launch: torchrun --nproc-per-node N --master-addr 127.0.0.1 examples/train_ddp.py
"""
import os
import sys
import time

import torch
import torch.nn as nn

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from torchdistpackage_amd import setup_distributed, tpc, fix_rand, NaiveDdp
from torchdistpackage_amd.models.gpt2 import GPT2Model, gpt2_small
from torchdistpackage_amd.ops.optim import FusedAdamW


def bench_wrapper(name, model, opt, x, steps=10, warmup=3, is_naive=False):
    def step():
        out = model(x, labels=x)
        out["loss"].backward()
        if is_naive:
            model.reduce_gradients()
        opt.step()
        opt.zero_grad()
    for _ in range(warmup):
        step()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        step()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / steps * 1e3
    print(f"{name}: {dt:.2f} ms/step")
    return dt


def main():
    info = setup_distributed()
    tpc.setup_process_groups([("data", info["world_size"])])
    fix_rand(info["rank"])
    dev = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
    dtype = torch.bfloat16 if dev.type == "cuda" else torch.float32
    cfg = gpt2_small()
    x = torch.randint(0, cfg.vocab_size, (8, 1024 if dev.type == "cuda" else 64),
                      device=dev)

    torch.manual_seed(0)
    m1 = GPT2Model(cfg, device=dev, dtype=dtype)
    naive = NaiveDdp(m1)
    bench_wrapper("NaiveDdp", naive, FusedAdamW(m1.parameters()), x,
                  is_naive=True)

    torch.manual_seed(0)
    m2 = GPT2Model(cfg, device=dev, dtype=dtype)
    if info["world_size"] > 1:
        m2 = nn.parallel.DistributedDataParallel(m2)
    bench_wrapper("TorchDDP", m2.module if hasattr(m2, "module") else m2,
                  FusedAdamW(m2.parameters()), x)


if __name__ == "__main__":
    main()
    import torch.distributed as _d
    if _d.is_initialized():
        _d.destroy_process_group()
