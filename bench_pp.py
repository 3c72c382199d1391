"""Pipeline-parallel path of the flagship bench: GPT-2 1.3B dp x pp x tp 1F1B.

Stage layers come from GPT2Model.to_stage_layers() ([embed, blocks..., head]);
stage boundaries carry SP activation shards (payload / tp) and the receiving
stage re-tags them.  NaiveDdp reduces only at the last micro-batch
(num_grad_acc_iter = num_microbatches) overlapped with the cooldown backwards.
"""

from __future__ import annotations

import time

import torch
import torch.distributed as dist
import torch.nn as nn


class PPStage(nn.Module):
    """One pipeline stage of a GPT2/Llama-style model."""

    def __init__(self, layers, is_first: bool, is_last: bool,
                 num_microbatches: int, tp_size: int):
        super().__init__()
        self.layers = nn.ModuleList(layers)
        self.is_first = is_first
        self.is_last = is_last
        self.num_mb = num_microbatches
        self.tp_size = tp_size

    def forward(self, inp, labels=None):
        from torchdistpackage_amd.parallel.tensor import (
            set_sequence_parallel_attr, is_sequence_parallel,
            gather_from_sequence_parallel_region)
        x = inp
        if not self.is_first and self.tp_size > 1:
            # stage boundaries carry SP shards (see bench_pp docstring)
            set_sequence_parallel_attr(x)
        head = None
        for layer in self.layers:
            from torchdistpackage_amd.models.gpt2 import GPT2Head
            if isinstance(layer, GPT2Head):
                head = layer
                # vocab-parallel heads gather SP internally (bwd
                # reduce-scatter); replicated heads need the pre-gather
                if not layer.vocab_parallel and \
                        is_sequence_parallel(x) and self.tp_size > 1:
                    x = gather_from_sequence_parallel_region(
                        x, bwd_mode="split")
            x = layer(x)
        if self.is_last:
            logits = x  # head output (B_mb, S, V) or (B_mb, S, V/tp)
            if head is not None and head.vocab_parallel:
                from torchdistpackage_amd.parallel.tensor.vocab import \
                    vocab_parallel_cross_entropy
                loss = vocab_parallel_cross_entropy(
                    logits.transpose(0, 1), labels.transpose(0, 1),
                    head.vocab_start, head.vocab_end) / self.num_mb
            else:
                from torchdistpackage_amd.ops import cross_entropy_loss
                loss = cross_entropy_loss(
                    logits.transpose(0, 1),
                    labels.transpose(0, 1)) / self.num_mb
            return loss
        return x


def run_pp_bench(args, cfg, dev, dtype, dp, pp, tp):
    from torchdistpackage_amd import tpc
    from torchdistpackage_amd.ddp import NaiveDdp
    from torchdistpackage_amd.models.gpt2 import GPT2Model
    from torchdistpackage_amd.ops.optim import FusedAdamW
    from torchdistpackage_amd.parallel.pipeline import (forward_backward,
                                                        partition_uniform)
    from torchdistpackage_amd.parallel.tensor import (
        allreduce_sequence_parallel_grads)

    cfg.tie_weights = False  # embed and head live on different stages
    torch.manual_seed(1234)
    full = GPT2Model(cfg)  # CPU fp32 init, stage extracted below
    layers = full.to_stage_layers()
    parts = partition_uniform(len(layers), pp)
    s, e = parts[tpc.get_pp_rank()]
    num_mb = args.micro_batches
    stage = PPStage(layers[s:e], tpc.is_first_in_pipeline_group(),
                    tpc.is_last_in_pipeline_group(), num_mb, tp)
    del full
    stage = stage.to(dev).to(dtype)

    if dp > 1:
        stage_ddp = NaiveDdp(stage, group=tpc.get_group("data"),
                             num_grad_acc_iter=num_mb)
    else:
        stage_ddp = stage
    opt = FusedAdamW(stage.parameters(), lr=1e-4, weight_decay=0.1)

    dp_rank = tpc.get_dp_rank()
    g = torch.Generator(device="cpu").manual_seed(9000 + dp_rank)
    B_total = args.batch * num_mb
    x = torch.randint(0, cfg.vocab_size, (B_total, args.seq),
                      generator=g).to(dev)

    def fwd_fn(stage_in, labels=None):
        # first stage receives the sliced token micro-batch, later stages
        # the previous stage's activation shard; the last stage gets its
        # label micro-batch from the scheduler (extra_inputs)
        m = stage_ddp if isinstance(stage_ddp, NaiveDdp) else stage
        return m(stage_in, labels=labels)

    def step():
        forward_backward(fwd_fn, inputs=x if stage.is_first else None,
                         num_microbatches=num_mb,
                         extra_inputs=x if stage.is_last else None)
        if isinstance(stage_ddp, NaiveDdp):
            stage_ddp.reduce_gradients()
        if tp > 1:
            allreduce_sequence_parallel_grads(stage)
        opt.step()
        opt.zero_grad()

    for _ in range(args.warmup):
        step()
    if dist.is_initialized():
        dist.barrier()
    if dev.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    if dev.type == "cuda":
        torch.cuda.synchronize()
    if dist.is_initialized():
        dist.barrier()
    dt = (time.perf_counter() - t0) / args.steps
    if dist.is_initialized():
        t = torch.tensor([dt], device=dev if dev.type == "cuda" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        dt = float(t.item())
    return {"ms_per_step": dt * 1e3}
