"""1F1B pipeline tests (gloo, CPU): numeric oracle vs single-process run.

Goes beyond the reference's smoke-only pipeline test
(examples/model_parallel/test_pipeline.py — "it runs"): grads after a full
1F1B iteration are compared against the same model run unpartitioned.
world_size=3 covers the mid-stage path the reference's pp=2 testing missed
(SURVEY.md known-bugs: pipeline_sched.py:129).
"""

import copy

import pytest
import torch
import torch.nn as nn

from tests.dist_helpers import run_distributed
from torchdistpackage_amd.parallel.pipeline.partition import (
    partition_balanced, partition_uniform, flatten_sequence)


def test_partition_uniform():
    assert partition_uniform(8, 2) == [[0, 4], [4, 8]]
    assert partition_uniform(7, 2) == [[0, 4], [4, 7]]
    assert partition_uniform(5, 4) == [[0, 2], [2, 3], [3, 4], [4, 5]]


def test_partition_balanced():
    layers = [nn.Linear(10, 10), nn.Linear(100, 100), nn.Linear(10, 10),
              nn.Linear(10, 10)]
    parts = partition_balanced(layers, 2)
    # the 100x100 layer dominates; it should sit alone-ish
    assert parts[0][1] == 2 and parts[1] == [2, 4]


def test_flatten_sequence():
    m = nn.Sequential(nn.Linear(2, 2),
                      nn.Sequential(nn.ReLU(), nn.Linear(2, 2)))
    flat = flatten_sequence(m)
    assert len(flat) == 3


def _make_layers(seed=0, depth=6, dim=32):
    torch.manual_seed(seed)
    return nn.Sequential(*[nn.Sequential(nn.Linear(dim, dim), nn.Tanh())
                           for _ in range(depth)])


def _pp_iteration(rank, world_size, num_microbatches=4):
    import torch.distributed as dist
    from torchdistpackage_amd.dist.topo import tpc
    from torchdistpackage_amd.parallel.pipeline import (forward_backward,
                                                        partition_uniform)

    tpc.setup_process_groups([("pipe", world_size)])
    depth, dim, B = 6, 32, 8
    full = _make_layers(seed=42, depth=depth, dim=dim)
    ref = copy.deepcopy(full)

    parts = partition_uniform(depth, world_size)
    s, e = parts[rank]
    stage = nn.Sequential(*list(full)[s:e])

    torch.manual_seed(7)
    x = torch.randn(B, dim)

    def fwd_fn(stage_in):
        out = stage(stage_in)
        if tpc.is_last_in_pipeline_group():
            # per-micro-batch loss normalized by num_microbatches
            return out.pow(2).mean() / num_microbatches
        return out

    losses = forward_backward(fwd_fn, inputs=x,
                              num_microbatches=num_microbatches,
                              return_losses=True)

    # oracle: same model, plain full-batch (mean over micro-batch losses ==
    # full-batch loss since micro-batches are equal-size)
    ref(x).pow(2).mean().backward()
    ref_stage = nn.Sequential(*list(ref)[s:e])
    for (n, p), (rn, rp) in zip(stage.named_parameters(),
                                ref_stage.named_parameters()):
        assert p.grad is not None, f"stage {rank} param {n} got no grad"
        assert torch.allclose(p.grad, rp.grad, atol=1e-5), \
            f"stage {rank} grad mismatch {n}: " \
            f"{(p.grad - rp.grad).abs().max().item()}"
    if tpc.is_last_in_pipeline_group():
        total = sum(float(l) for l in losses)
        ref_loss = float(ref(x).pow(2).mean())
        assert abs(total - ref_loss) < 1e-5
    return True


def test_1f1b_pp2():
    run_distributed(_pp_iteration, world_size=2)


def test_1f1b_pp3():
    run_distributed(_pp_iteration, world_size=3)


def test_1f1b_pp2_many_microbatches():
    run_distributed(_pp_iteration, world_size=2,
                    kwargs={"num_microbatches": 8})


def _pp_eval(rank, world_size):
    import torch.distributed as dist
    from torchdistpackage_amd.dist.topo import tpc
    from torchdistpackage_amd.parallel.pipeline import (forward_eval,
                                                        partition_uniform)

    tpc.setup_process_groups([("pipe", world_size)])
    depth, dim, B = 4, 16, 4
    full = _make_layers(seed=3, depth=depth, dim=dim)
    parts = partition_uniform(depth, world_size)
    s, e = parts[rank]
    stage = nn.Sequential(*list(full)[s:e])
    torch.manual_seed(5)
    x = torch.randn(B, dim)
    outs = forward_eval(lambda t: stage(t), inputs=x, num_microbatches=2)
    if tpc.is_last_in_pipeline_group():
        with torch.no_grad():
            ref = full(x)
        got = torch.cat(outs, dim=0)
        assert torch.allclose(got, ref, atol=1e-6)
    return True


def test_forward_eval_pp2():
    run_distributed(_pp_eval, world_size=2)


def _pp_with_ddp(rank, world_size):
    """dp2 x pp2 on 4 ranks: NaiveDdp with num_grad_acc_iter=num_microbatches
    reduces only at the last micro-batch (reference Readme.md:56 claim)."""
    import torch.distributed as dist
    from torchdistpackage_amd.dist.topo import tpc
    from torchdistpackage_amd.ddp import NaiveDdp
    from torchdistpackage_amd.parallel.pipeline import (forward_backward,
                                                        partition_uniform)

    tpc.setup_process_groups([("data", 2), ("pipe", 2)])
    depth, dim, B, n_mb = 4, 16, 8, 4
    full = _make_layers(seed=11, depth=depth, dim=dim)
    ref = copy.deepcopy(full)
    s, e = partition_uniform(depth, 2)[tpc.get_pp_rank()]
    stage = nn.Sequential(*list(full)[s:e])
    stage_ddp = NaiveDdp(stage, group=tpc.get_group("data"),
                         num_grad_acc_iter=n_mb)

    dp_rank = tpc.get_dp_rank()
    torch.manual_seed(100 + dp_rank)
    x = torch.randn(B, dim)

    def fwd_fn(stage_in):
        out = stage_ddp(stage_in)
        if tpc.is_last_in_pipeline_group():
            return out.pow(2).mean() / n_mb
        return out

    forward_backward(fwd_fn, inputs=x, num_microbatches=n_mb)
    stage_ddp.reduce_gradients()

    # oracle: average of both dp ranks' full-batch grads
    for r in range(2):
        torch.manual_seed(100 + r)
        xr = torch.randn(B, dim)
        ref(xr).pow(2).mean().backward()
    ref_stage = nn.Sequential(*list(ref)[s:e])
    for (n, p), (rn, rp) in zip(stage.named_parameters(),
                                ref_stage.named_parameters()):
        assert torch.allclose(p.grad, rp.grad / 2, atol=1e-5), \
            f"{n}: {(p.grad - rp.grad / 2).abs().max().item()}"
    return True


def test_pp2_dp2_composition():
    run_distributed(_pp_with_ddp, world_size=4)


def _pp_extra_inputs(rank, world_size, num_microbatches=4):
    """Scheduler-native extra per-stage inputs (reference
    pipeline_sched.py:6-33): last-stage labels sliced into micro-batches by
    forward_backward itself — no mutable stage-state side channel."""
    from torchdistpackage_amd.dist.topo import tpc
    from torchdistpackage_amd.parallel.pipeline import (forward_backward,
                                                        partition_uniform)

    tpc.setup_process_groups([("pipe", world_size)])
    depth, dim, B = 6, 32, 8
    full = _make_layers(seed=42, depth=depth, dim=dim)
    ref = copy.deepcopy(full)

    parts = partition_uniform(depth, world_size)
    s, e = parts[rank]
    stage = nn.Sequential(*list(full)[s:e])

    torch.manual_seed(7)
    x = torch.randn(B, dim)
    labels = torch.randn(B, dim)
    is_last = tpc.is_last_in_pipeline_group()

    seen_mb = []

    def fwd_fn(stage_in, extra=None):
        out = stage(stage_in)
        if is_last:
            assert extra is not None and extra.shape[0] == B // num_microbatches
            seen_mb.append(extra)
            return (out - extra).pow(2).mean() / num_microbatches
        return out

    forward_backward(fwd_fn, inputs=x, num_microbatches=num_microbatches,
                     extra_inputs=labels if is_last else None)

    if is_last:
        # micro-batches must arrive in order and tile the full labels
        assert torch.equal(torch.cat(seen_mb, dim=0), labels)

    (ref(x) - labels).pow(2).mean().backward()
    ref_stage = nn.Sequential(*list(ref)[s:e])
    for (n, p), (rn, rp) in zip(stage.named_parameters(),
                                ref_stage.named_parameters()):
        assert p.grad is not None
        assert torch.allclose(p.grad, rp.grad, atol=1e-5), (rank, n)
    return True


def test_1f1b_extra_inputs_pp2():
    run_distributed(_pp_extra_inputs, world_size=2)


def test_1f1b_extra_inputs_pp3():
    run_distributed(_pp_extra_inputs, world_size=3)


def _pp_tp_vocab_parallel_parity(rank, world_size):
    """The 8-GPU flagship composition in miniature (pp2 x tp2, gloo):
    GPT-2 stages with the vocab-parallel embedding/head/CE across PP
    boundaries must reproduce the tp=1/pp=1 oracle loss (weight surgery)."""
    import torch.distributed as dist
    from torchdistpackage_amd.dist.topo import tpc
    from torchdistpackage_amd.parallel.tensor import set_tp_group
    from torchdistpackage_amd.models.gpt2 import GPT2Config, GPT2Model
    from torchdistpackage_amd.parallel.pipeline import (forward_backward,
                                                        partition_uniform)
    import sys, os
    sys.path.insert(0, os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    from bench_pp import PPStage

    cfg = GPT2Config(vocab_size=128, n_layer=2, n_head=4, dim=32, max_seq=16,
                     tie_weights=False)
    torch.manual_seed(0)
    oracle = GPT2Model(cfg)       # before TP groups exist: tp=1 semantics
    torch.manual_seed(2)
    x = torch.randint(0, 128, (4, 16))
    loss_ref = float(oracle(x, labels=x)["loss"])

    tpc.setup_process_groups([("pipe", 2), ("tensor", 2)])
    set_tp_group(tpc.get_group("tensor"))
    torch.manual_seed(0)
    tp_model = GPT2Model(cfg)
    # surgery: oracle weights -> TP shards
    tp_model.embed.wte.load_from_full(oracle.embed.wte.weight)
    tp_model.embed.wpe.load_state_dict(oracle.embed.wpe.state_dict())
    tp_model.head.ln_f.load_state_dict(oracle.head.ln_f.state_dict())
    with torch.no_grad():
        tp_model.head.weight.copy_(
            oracle.head.weight[tp_model.head.vocab_start:
                               tp_model.head.vocab_end])
    for fb, tb in zip(oracle.blocks, tp_model.blocks):
        tb.init_from_full(fb)

    layers = tp_model.to_stage_layers()
    parts = partition_uniform(len(layers), 2)
    s, e = parts[tpc.get_pp_rank()]
    num_mb = 2
    stage = PPStage(layers[s:e], tpc.is_first_in_pipeline_group(),
                    tpc.is_last_in_pipeline_group(), num_mb, 2)

    def fwd_fn(stage_in, labels=None):
        return stage(stage_in, labels=labels)

    losses = forward_backward(
        fwd_fn, inputs=x if stage.is_first else None,
        num_microbatches=num_mb,
        extra_inputs=x if stage.is_last else None,
        return_losses=True)
    if stage.is_last:
        total = sum(float(l) for l in losses)
        assert abs(total - loss_ref) < 2e-3, (total, loss_ref)
    # every stage's params must have grads (except frozen none here)
    for n, p in stage.named_parameters():
        assert p.grad is not None, n
    return True


def test_pp2_tp2_vocab_parallel_parity():
    run_distributed(_pp_tp_vocab_parallel_parity, world_size=4)


def test_1f1b_pp4():
    """Deeper pipe: 3 mid stages exercise warmup depth 3 + fused steady
    paths (the reference's pp>=3 bug class)."""
    run_distributed(_pp_iteration, world_size=4,
                    kwargs={"num_microbatches": 8})
