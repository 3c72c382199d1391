"""Tensor + sequence parallel GPT-2 training (reference
examples/model_parallel/test_transformer.py composition, plus the
vocab-parallel embedding/head/CE this package adds).

Launch: torchrun --nproc-per-node 2 --master-addr 127.0.0.1 examples/train_tp.py
"""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from torchdistpackage_amd import setup_distributed, tpc, fix_rand
from torchdistpackage_amd.models.gpt2 import GPT2Config, GPT2Model
from torchdistpackage_amd.ops.optim import FusedAdamW
from torchdistpackage_amd.parallel.tensor import (
    set_tp_group, allreduce_sequence_parallel_grads)


def main():
    info = setup_distributed()
    tp = info["world_size"]
    tpc.setup_process_groups([("tensor", tp)])
    set_tp_group(tpc.get_group("tensor"))
    fix_rand(0)   # identical shards everywhere (vocab-parallel tables are
    #               drawn full-then-sliced, TP weights per-shard)
    dev = torch.device("cuda") if torch.cuda.is_available() \
        else torch.device("cpu")
    dtype = torch.bfloat16 if dev.type == "cuda" else torch.float32

    cfg = GPT2Config(vocab_size=1024, n_layer=4, n_head=8, dim=256,
                     max_seq=256, sequence_parallel=True)
    model = GPT2Model(cfg, device=dev, dtype=dtype)
    opt = FusedAdamW(model.parameters(), lr=3e-4)

    fix_rand(7)   # same tokens on every TP rank (they share the batch)
    x = torch.randint(0, cfg.vocab_size, (8, 256), device=dev)
    for it in range(5):
        loss = model(x, labels=x)["loss"]
        loss.backward()
        # SP-region params (LayerNorm, wpe) hold shard-local grads
        allreduce_sequence_parallel_grads(model)
        opt.step()
        opt.zero_grad()
        if info["rank"] == 0:
            print(f"iter {it} loss {loss.item():.4f}")


if __name__ == "__main__":
    main()
    import torch.distributed as _d
    if _d.is_initialized():
        _d.destroy_process_group()
