"""MoE training: expert-parallel all-to-all + MoE-DP replicas.

Reference gap filled: the reference delegates EP to DeepSpeed
(explore/moe/ds_fmoe_main.py); here the whole stack is in-package.

Launch: torchrun --nproc-per-node N --master-addr 127.0.0.1 examples/train_moe.py
"""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from torchdistpackage_amd import (setup_distributed, tpc, fix_rand, NaiveDdp,
                                  create_moe_dp_hooks, moe_dp_iter_step)
from torchdistpackage_amd.models.moe_model import MoEConfig, MoEModel
from torchdistpackage_amd.ops.optim import FusedAdamW


def main():
    info = setup_distributed()
    ws = info["world_size"]
    ep = min(ws, 4)
    tpc.setup_process_groups([("data", ws)])
    tpc.build_moe_groups(moe_dp_size=ws // ep, moe_ep_size=ep)
    fix_rand(info["rank"])
    dev = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
    dtype = torch.bfloat16 if dev.type == "cuda" else torch.float32

    cfg = MoEConfig(vocab_size=1024, n_layer=2, n_head=4, dim=256,
                    max_seq=128, num_experts=ep * 2, top_k=2)
    model = MoEModel(cfg, device=dev, dtype=dtype)
    # dense params sync over 'data'; expert params over 'moe_dp'
    ddp = NaiveDdp(model)
    if ws // ep > 1:
        create_moe_dp_hooks(list(model.expert_parameters()))
    opt = FusedAdamW(model.parameters(), lr=1e-4)

    for it in range(3):
        x = torch.randint(0, cfg.vocab_size, (2, 128), device=dev)
        out = ddp(x, labels=x)
        out["loss"].backward()
        ddp.reduce_gradients()
        moe_dp_iter_step()
        opt.step()
        opt.zero_grad()
        if info["rank"] == 0:
            print(f"iter {it} loss {out['loss'].item():.4f} "
                  f"aux {out['aux_loss'].item():.4f}")


if __name__ == "__main__":
    main()
    import torch.distributed as _d
    if _d.is_initialized():
        _d.destroy_process_group()
