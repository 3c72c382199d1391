"""ZeRO-1/2 + hybrid node groups + sharded EMA (reference test_zero_optim.py
composition, Llama backbone, BASELINE config 5 shape).

Launch: torchrun --nproc-per-node N --master-addr 127.0.0.1 examples/train_zero.py
"""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from torchdistpackage_amd import (setup_distributed, tpc, fix_rand,
                                  Bf16ZeroOptimizer, ShardedEMA,
                                  setup_node_groups, save_checkpoint)
from torchdistpackage_amd.models.llama import LlamaModel, llama_tiny
from torchdistpackage_amd.ops.optim import FusedAdamW


def main():
    info = setup_distributed()
    tpc.setup_process_groups([("data", info["world_size"])])
    node_group = setup_node_groups(
        num_per_node=min(8, info["world_size"]))
    # seed IDENTICALLY for model init so DP replicas start from the same
    # weights (ZeRO never broadcasts params); re-seed per-rank for data below
    fix_rand(0)
    dev = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
    dtype = torch.bfloat16 if dev.type == "cuda" else torch.float32

    model = LlamaModel(llama_tiny(), device=dev, dtype=dtype)
    fix_rand(info["rank"])
    inner = FusedAdamW(model.parameters(), lr=1e-4)
    # hybrid: shard optimizer state inside the node, reduce grads over all DP
    opt = Bf16ZeroOptimizer(inner, group=node_group,
                            grad_group=tpc.get_group("data"), stage2=True,
                            clip_grad=1.0)
    ema = ShardedEMA(model, decay=0.999)

    for it in range(5):
        x = torch.randint(0, 512, (4, 128), device=dev)
        loss = model(x, labels=x)["loss"]
        loss.backward()
        opt.step()
        opt.zero_grad()
        ema.update()
        if info["rank"] == 0:
            print(f"iter {it} loss {loss.item():.4f}")
    save_checkpoint("/tmp/tdpa_zero_ckpt", 5, model, ema=ema)


if __name__ == "__main__":
    main()
    import torch.distributed as _d
    if _d.is_initialized():
        _d.destroy_process_group()
