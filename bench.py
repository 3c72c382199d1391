"""Flagship benchmark: GPT-2 1.3B training step, tokens/sec whole-node.

BASELINE.json metric: "tokens/sec (whole node) GPT-2-1.3B DP+TP2+PP2 at
1/2/4/8 MI355X".  Parallelism by GPU count (dist_config ordered
[data, pipe, tensor], tensor innermost):

    N=1: plain single GPU          N=2: dp2
    N=4: dp4                       N=8: [('data',2),('pipe',2),('tensor',2)]

Launched by the driver as
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W
Prints ONE JSON line from rank 0 (whole-job aggregate tokens/s).
Synthetic data (random token ids), random-init weights, bf16 compute,
fp32 master weights + fused HIP AdamW.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


def parse_args():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--batch", type=int, default=16,
                    help="micro/global batch per DP rank")
    ap.add_argument("--seq", type=int, default=1024)
    ap.add_argument("--model", type=str, default="gpt2_1.3b",
                    choices=["gpt2_1.3b", "gpt2_small", "tiny", "llama_8b",
                             "moe_8x", "moe_tiny", "llama_tiny"])
    ap.add_argument("--no-graph", action="store_true",
                    help="disable the small-model hipGraph default")
    ap.add_argument("--graph", action="store_true",
                    help="capture the whole train step in one hipGraph "
                         "(world_size==1 only)")
    ap.add_argument("--zero", action="store_true",
                    help="use Bf16ZeroOptimizer (hybrid node-local shard) + "
                         "sharded EMA instead of plain FusedAdamW "
                         "(dp/tp path only; ignored when pp > 1)")
    ap.add_argument("--micro-batches", type=int, default=8,
                    help="micro-batches per step when PP is active")
    return ap.parse_args()


# dp scales best below 8 GPUs (grad all-reduce overlaps with backward;
# TP/SP collectives sit on the critical path); the 8-GPU layout is the
# BASELINE-named dp2 x pp2 x tp2.
PARALLEL_MAP = {
    1: (1, 1, 1),
    2: (2, 1, 1),
    4: (4, 1, 1),
    8: (2, 2, 2),
}


def main():
    args = parse_args()
    from torchdistpackage_amd import setup_distributed, tpc, fix_rand
    from torchdistpackage_amd.ddp import NaiveDdp
    from torchdistpackage_amd.models.gpt2 import (GPT2Config, GPT2Model,
                                                  gpt2_small, gpt2_xl_1p3b)
    from torchdistpackage_amd.ops.optim import FusedAdamW

    world = int(os.environ.get("WORLD_SIZE", 1))
    if world > 1:
        info = setup_distributed()
        rank = info["rank"]
    else:
        rank = 0
        if torch.cuda.is_available():
            torch.cuda.set_device(0)
    import torch.distributed as dist

    if os.environ.get("TDPA_PARALLEL"):   # "dp,pp,tp" override for testing
        dp, pp, tp = (int(v) for v in os.environ["TDPA_PARALLEL"].split(","))
    else:
        dp, pp, tp = PARALLEL_MAP.get(world, (world, 1, 1))
    if world > 1:
        tpc.setup_process_groups(
            [("data", dp), ("pipe", pp), ("tensor", tp)])
    use_pp = pp > 1
    if args.zero and use_pp:
        raise SystemExit("--zero is wired for the dp(/tp) layouts here; "
                         "with pp>1 each stage holds different params and "
                         "the bench's shard-group selection does not apply "
                         "(compose Bf16ZeroOptimizer per stage directly)")

    fix_rand(tpc.get_dp_rank() if world > 1 else 0)

    if args.model == "gpt2_1.3b":
        cfg = gpt2_xl_1p3b()
    elif args.model == "gpt2_small":
        cfg = gpt2_small()
    elif args.model == "llama_8b":
        from torchdistpackage_amd.models.llama import llama3_8b
        cfg = llama3_8b()
    elif args.model == "moe_8x":
        from torchdistpackage_amd.models.moe_model import mixtral_style_8x
        cfg = mixtral_style_8x()
    elif args.model == "llama_tiny":
        from torchdistpackage_amd.models.llama import llama_tiny
        cfg = llama_tiny()
    elif args.model == "moe_tiny":
        from torchdistpackage_amd.models.moe_model import MoEConfig
        cfg = MoEConfig(vocab_size=512, n_layer=2, n_head=2, dim=128,
                        max_seq=args.seq, num_experts=4, top_k=2,
                        hidden_mult=2)
    else:
        cfg = GPT2Config(vocab_size=2048, n_layer=4, n_head=8, dim=512,
                         max_seq=args.seq)
    cfg.max_seq = max(cfg.max_seq, args.seq)

    dev = torch.device("cuda", torch.cuda.current_device()) \
        if torch.cuda.is_available() else torch.device("cpu")
    dtype = torch.bfloat16 if dev.type == "cuda" else torch.float32

    if use_pp:
        from bench_pp import run_pp_bench  # PP path in separate module
        result = run_pp_bench(args, cfg, dev, dtype, dp, pp, tp)
    else:
        result = run_dp_tp_bench(args, cfg, dev, dtype, dp, tp)

    if rank == 0:
        B_global = args.batch * dp * args.micro_batches if use_pp \
            else args.batch * dp
        tokens_per_step = B_global * args.seq
        toks_per_s = tokens_per_step / (result["ms_per_step"] / 1e3)
        print(json.dumps({
            "metric": "tokens_per_second",
            "value": toks_per_s,
            "unit": "tokens/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": result["ms_per_step"],
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if dtype == torch.bfloat16 else "fp32",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": B_global,
                "seq_len": args.seq,
                "parallelism": f"dp{dp}_pp{pp}_tp{tp}",
                **({"torchddp_ms_per_step":
                    round(result["torchddp_ms_per_step"], 3),
                    "naive_vs_torchddp_speedup":
                    round(result["naive_vs_torchddp_speedup"], 4)}
                   if "torchddp_ms_per_step" in result else {}),
            },
        }), flush=True)
    if world > 1:
        dist.destroy_process_group()


def run_dp_tp_bench(args, cfg, dev, dtype, dp, tp):
    import torch.distributed as dist
    from torchdistpackage_amd import tpc
    from torchdistpackage_amd.ddp import NaiveDdp
    from torchdistpackage_amd.models.gpt2 import GPT2Model
    from torchdistpackage_amd.ops.optim import FusedAdamW

    world = dist.get_world_size() if dist.is_initialized() else 1
    torch.manual_seed(1234)  # same init across ranks (then broadcast anyway)
    if args.model.startswith("llama"):
        from torchdistpackage_amd.models.llama import LlamaModel
        model = LlamaModel(cfg, device=dev, dtype=dtype)
    elif args.model.startswith("moe"):
        from torchdistpackage_amd.models.moe_model import MoEModel
        if world > 1:
            # groups must exist BEFORE the model captures its EP group
            ep = min(dp, cfg.num_experts)
            tpc.build_moe_groups(moe_dp_size=dp // ep, moe_ep_size=ep)
        model = MoEModel(cfg, device=dev, dtype=dtype)
    else:
        model = GPT2Model(cfg, device=dev, dtype=dtype)

    ema = None
    if args.zero and args.model.startswith("moe"):
        raise SystemExit("--zero shards optimizer state over the dp group, "
                         "which is wrong for expert-parallel params (each EP "
                         "rank owns different experts); use NaiveDdp+MoeDP")

    if args.zero:
        # ZeRO owns grad reduction — model stays unwrapped (wrapping with
        # NaiveDdp too would leave BOTH hook sets firing).
        # Sharding must stay WITHIN a set of ranks holding IDENTICAL params:
        # the intra-node hybrid group only when the node is pure-dp; with
        # tp/pp active only the dp group is replicated (an all-gather over
        # mixed tp/pp ranks would overwrite each shard's weights).
        from torchdistpackage_amd import Bf16ZeroOptimizer, ShardedEMA, \
            setup_node_groups
        if world > 1 and tp > 1:   # this path runs with pp == 1
            shard_group = tpc.get_group("data")
        elif world > 1:
            shard_group = setup_node_groups(num_per_node=min(8, world))
        else:
            shard_group = None
        inner_opt = FusedAdamW(model.parameters(), lr=1e-4, weight_decay=0.1)
        opt = Bf16ZeroOptimizer(inner_opt, group=shard_group,
                                grad_group=tpc.get_group("data")
                                if world > 1 else None, stage2=True)
        ema = ShardedEMA(model, decay=0.999,
                         group=tpc.get_group("data") if world > 1 else None)
    else:
        if world > 1 and dp > 1:
            model = NaiveDdp(model, group=tpc.get_group("data"))
            if args.model.startswith("moe"):
                from torchdistpackage_amd.ddp import create_moe_dp_hooks
                inner = model.module
                if tpc.get_group_size("moe_dp") > 1:
                    create_moe_dp_hooks(list(inner.expert_parameters()))
        opt = FusedAdamW(model.parameters(), lr=1e-4, weight_decay=0.1)

    # identical data inside a TP group; different across DP ranks
    dp_rank = tpc.get_dp_rank() if world > 1 else 0
    g = torch.Generator(device="cpu").manual_seed(9000 + dp_rank)
    x = torch.randint(0, cfg.vocab_size, (args.batch, args.seq),
                      generator=g).to(dev)

    from torchdistpackage_amd.parallel.tensor import \
        allreduce_sequence_parallel_grads

    def step():
        out = model(x, labels=x)
        out["loss"].backward()
        if isinstance(model, NaiveDdp):
            model.reduce_gradients()
            from torchdistpackage_amd.ddp import moe_dp_iter_step
            if args.model.startswith("moe"):
                moe_dp_iter_step()
        if tp > 1:
            allreduce_sequence_parallel_grads(
                model.module if isinstance(model, NaiveDdp) else model)
        opt.step()
        opt.zero_grad()
        if ema is not None:
            ema.update()

    stepper = step
    # hipGraph capture: default ON for the launch-bound small models
    # (gpt2_small same-box: 450k tok/s graphed vs 340k eager, r02); the
    # GPU-bound 1.3B shape measured ~5% SLOWER graphed (r01 v14), so big
    # models stay eager unless --graph is passed.
    use_graph = (args.graph or args.model in ("gpt2_small", "tiny")) \
        and not args.no_graph
    if use_graph and world == 1 and dev.type == "cuda" and ema is None:
        # hipGraph capture: grads must keep stable storage
        from torchdistpackage_amd.utils_graph import GraphedStep

        def graph_step():
            out = model(x, labels=x)
            out["loss"].backward()
            opt.step()
            opt.zero_grad(set_to_none=False)

        gs = GraphedStep(graph_step, warmup=max(args.warmup, 3))
        stepper = gs.replay

    for _ in range(args.warmup):
        stepper()
    if dist.is_initialized():
        dist.barrier()
    if dev.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        stepper()
    if dev.type == "cuda":
        torch.cuda.synchronize()
    if dist.is_initialized():
        dist.barrier()
    dt = (time.perf_counter() - t0) / args.steps
    # max over ranks
    if dist.is_initialized():
        t = torch.tensor([dt], device=dev if dev.type == "cuda" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        dt = float(t.item())
    result = {"ms_per_step": dt * 1e3}

    # --- NaiveDdp vs TorchDDP step time (headline sub-metric) ----------
    # best-effort: a failure here must never kill the main bench result
    try:
        _compare_torchddp(args, model, dev, dp, x, result)
    except Exception as e:  # noqa: BLE001
        print(f"[bench] torchddp comparison skipped: {e}", file=sys.stderr)
    return result


def _compare_torchddp(args, model, dev, dp, x, result):
    import time
    import torch.distributed as dist
    from torchdistpackage_amd import tpc
    from torchdistpackage_amd.ddp import NaiveDdp
    from torchdistpackage_amd.ops.optim import FusedAdamW
    if isinstance(model, NaiveDdp) and dp > 1 and dev.type == "cuda" \
            and args.model.startswith("gpt2"):
        import torch.nn as nn
        model.remove_hooks()
        tddp = nn.parallel.DistributedDataParallel(
            model.module, process_group=tpc.get_group("data"))
        opt2 = FusedAdamW(tddp.parameters(), lr=1e-4, weight_decay=0.1)

        def step2():
            out = tddp(x, labels=x)
            out["loss"].backward()
            opt2.step()
            opt2.zero_grad()

        for _ in range(args.warmup):
            step2()
        dist.barrier()
        torch.cuda.synchronize()
        t0b = time.perf_counter()
        for _ in range(args.steps):
            step2()
        torch.cuda.synchronize()
        dist.barrier()
        dtb = (time.perf_counter() - t0b) / args.steps
        tb = torch.tensor([dtb], device=dev)
        dist.all_reduce(tb, op=dist.ReduceOp.MAX)
        result["torchddp_ms_per_step"] = float(tb.item()) * 1e3
        result["naive_vs_torchddp_speedup"] = \
            result["torchddp_ms_per_step"] / result["ms_per_step"]


if __name__ == "__main__":
    main()
