"""1F1B pipeline training (reference examples/model_parallel/test_pipeline.py
composition): GPT-2 partitioned over pp stages + DP via NaiveDdp with
last-micro-batch-only reduce.

Launch: TDPA_PARALLEL=1,2,1 torchrun --nproc-per-node 2 --master-addr 127.0.0.1 examples/train_pipeline.py
"""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from torchdistpackage_amd import setup_distributed, tpc, fix_rand
from torchdistpackage_amd.models.gpt2 import GPT2Config
from bench_pp import run_pp_bench  # reuse the canonical PP composition


class Args:
    batch = 2
    seq = 128
    micro_batches = 4
    steps = 3
    warmup = 1


def main():
    info = setup_distributed()
    dp, pp, tp = (int(v) for v in os.environ.get(
        "TDPA_PARALLEL", f"1,{info['world_size']},1").split(","))
    tpc.setup_process_groups([("data", dp), ("pipe", pp), ("tensor", tp)])
    fix_rand(tpc.get_dp_rank())
    dev = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
    dtype = torch.bfloat16 if dev.type == "cuda" else torch.float32
    cfg = GPT2Config(vocab_size=1024, n_layer=4, n_head=4, dim=256,
                     max_seq=Args.seq)
    res = run_pp_bench(Args, cfg, dev, dtype, dp, pp, tp)
    if info["rank"] == 0:
        print(f"pp bench: {res['ms_per_step']:.2f} ms/step")


if __name__ == "__main__":
    main()
    import torch.distributed as _d
    if _d.is_initialized():
        _d.destroy_process_group()
