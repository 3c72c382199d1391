"""Bf16ZeroOptimizer differential test vs DDP+Adam (gloo world_size=2, CPU).

Mirrors the reference's examples/test_zero_optim.py oracle method.
"""

import copy

import torch
import torch.nn as nn

from tests.dist_helpers import run_distributed


def _make_model(seed=0):
    torch.manual_seed(seed)
    return nn.Sequential(nn.Linear(48, 96), nn.Tanh(), nn.Linear(96, 48),
                         nn.Tanh(), nn.Linear(48, 8))


def _zero_vs_ddp(rank, world_size, stage2=False):
    from torchdistpackage_amd.ddp import Bf16ZeroOptimizer

    model_a = _make_model(seed=11)
    model_b = copy.deepcopy(model_a)

    inner = torch.optim.Adam(model_a.parameters(), lr=1e-3)
    zopt = Bf16ZeroOptimizer(inner, stage2=stage2)

    ref = nn.parallel.DistributedDataParallel(model_b)
    opt_b = torch.optim.Adam(model_b.parameters(), lr=1e-3)

    for it in range(5):
        torch.manual_seed(500 + 10 * it + rank)
        x = torch.randn(6, 48)

        model_a(x).pow(2).mean().backward()
        ref(x).pow(2).mean().backward()

        zopt.step()
        opt_b.step()
        zopt.zero_grad()
        opt_b.zero_grad()

        for (na, pa), (nb, pb) in zip(model_a.named_parameters(),
                                      model_b.named_parameters()):
            assert torch.allclose(pa, pb, atol=1e-5), \
                f"iter {it} param {na}: max diff " \
                f"{(pa - pb).abs().max().item()}"
    return True


def test_zero1():
    run_distributed(_zero_vs_ddp, world_size=2)


def test_zero2():
    run_distributed(_zero_vs_ddp, world_size=2, kwargs={"stage2": True})


def _zero_single_rank(rank, world_size):
    # world_size=1 path: master-grad copy without comm
    from torchdistpackage_amd.ddp import Bf16ZeroOptimizer

    model = _make_model(seed=2)
    ref = copy.deepcopy(model)
    zopt = Bf16ZeroOptimizer(torch.optim.Adam(model.parameters(), lr=1e-3))
    opt_ref = torch.optim.Adam(ref.parameters(), lr=1e-3)
    for it in range(3):
        torch.manual_seed(it)
        x = torch.randn(4, 48)
        model(x).sum().backward()
        ref(x).sum().backward()
        zopt.step()
        opt_ref.step()
        zopt.zero_grad()
        opt_ref.zero_grad()
    for pa, pb in zip(model.parameters(), ref.parameters()):
        assert torch.allclose(pa, pb, atol=1e-6)
    return True


def test_zero_world1():
    run_distributed(_zero_single_rank, world_size=1)


def _zero_clip(rank, world_size):
    from torchdistpackage_amd.ddp import Bf16ZeroOptimizer
    model = _make_model(seed=5)
    zopt = Bf16ZeroOptimizer(torch.optim.Adam(model.parameters(), lr=1e-3),
                             clip_grad=0.1)
    torch.manual_seed(rank)
    (model(torch.randn(4, 48) * 100).pow(2).mean()).backward()
    zopt.step()  # just verifies the clip path runs distributed
    return True


def test_zero_clip_grad():
    run_distributed(_zero_clip, world_size=2)


def _zero_with_fused_adamw(rank, world_size):
    """Composition regression: ZeRO's master views have requires_grad=False
    and must still be updated by FusedAdamW (a silent no-op bug once)."""
    from torchdistpackage_amd.ddp import Bf16ZeroOptimizer
    from torchdistpackage_amd.ops.optim import FusedAdamW

    model = _make_model(seed=21)
    opt = Bf16ZeroOptimizer(FusedAdamW(model.parameters(), lr=1e-2))
    p0 = next(model.parameters()).detach().clone()
    losses = []
    torch.manual_seed(3)
    x = torch.randn(4, 48)
    for it in range(5):
        loss = model(x).pow(2).mean()
        loss.backward()
        opt.step()
        opt.zero_grad()
        losses.append(loss.item())
    assert not torch.equal(p0, next(model.parameters())), "params frozen"
    assert losses[-1] < losses[0], f"loss flat: {losses}"
    return True


def test_zero_with_fused_adamw():
    run_distributed(_zero_with_fused_adamw, world_size=1)


def test_zero_with_fused_adamw_world2():
    run_distributed(_zero_with_fused_adamw, world_size=2)


def _zero_tp_sp_grads(rank, world_size):
    """ADVICE r01 regression: ZeRO + TP/SP composition.  SP-tagged params'
    grads are fixed up by an all-reduce over TP AFTER backward; ZeRO must not
    snapshot them into buckets before that (the bucketed copy silently
    dropped the TP all-reduce)."""
    from torchdistpackage_amd.dist.topo import tpc
    from torchdistpackage_amd.ddp import Bf16ZeroOptimizer
    from torchdistpackage_amd.parallel.tensor.tp_utils import (
        set_tp_group, allreduce_sequence_parallel_grads)

    tpc.setup_process_groups([("data", 2), ("tensor", 2)])
    set_tp_group(tpc.get_group("tensor"))
    dp_rank = tpc.get_dp_rank()
    tp_rank = tpc.get_tp_rank()

    torch.manual_seed(7)
    model = nn.Linear(8, 8, bias=True)
    model.bias.sequence_parallel_param = True  # plays the SP LayerNorm role

    inner = torch.optim.Adam(model.parameters(), lr=1e-3)
    zopt = Bf16ZeroOptimizer(inner, group=tpc.get_group("data"),
                             grad_group=tpc.get_group("data"))

    # deterministic per-rank grads: d(loss)/dw = a per element, d/db = c
    a = 1.0 + dp_rank
    c = 1.0 + 2 * dp_rank + 10 * tp_rank
    (model.weight.sum() * a + model.bias.sum() * c).backward()
    allreduce_sequence_parallel_grads(model)
    zopt._finish_reduction()

    # expected: weight grad = avg_dp(a) = 1.5;
    # bias grad = avg_dp(sum_tp(c)) = ((12+0)+(12+4))/2 = 14
    for i in zopt._my_idx:
        mp = zopt._master_params[i]
        p = zopt._params[i]
        expect = 1.5 if p is model.weight else 14.0
        assert mp.grad is not None
        assert torch.allclose(mp.grad, torch.full_like(mp, expect)), \
            (rank, p.shape, mp.grad.flatten()[:4], expect)
    return True


def test_zero_tp_sp_grads_world4():
    run_distributed(_zero_tp_sp_grads, world_size=4)


def _zero_grad_acc(rank, world_size):
    """num_grad_acc_iter: two accumulated backwards per step must match one
    full-batch step (DDP+Adam oracle), with no intermediate reduces."""
    from torchdistpackage_amd.ddp import Bf16ZeroOptimizer

    model_a = _make_model(seed=21)
    model_b = copy.deepcopy(model_a)
    inner = torch.optim.Adam(model_a.parameters(), lr=1e-3)
    zopt = Bf16ZeroOptimizer(inner, num_grad_acc_iter=2)
    ref = nn.parallel.DistributedDataParallel(model_b)
    opt_b = torch.optim.Adam(model_b.parameters(), lr=1e-3)

    for it in range(3):
        torch.manual_seed(900 + 10 * it + rank)
        x = torch.randn(8, 48)
        # accumulated halves (scaled so grads match the full batch)
        (model_a(x[:4]).pow(2).mean() / 2).backward()
        (model_a(x[4:]).pow(2).mean() / 2).backward()
        (ref(x[:4]).pow(2).mean() / 2).backward()
        (ref(x[4:]).pow(2).mean() / 2).backward()
        zopt.step()
        opt_b.step()
        zopt.zero_grad()
        opt_b.zero_grad()
        for (na, pa), (nb, pb) in zip(model_a.named_parameters(),
                                      model_b.named_parameters()):
            assert torch.allclose(pa, pb, atol=1e-5), (it, na)
    return True


def test_zero_grad_accumulation():
    run_distributed(_zero_grad_acc, world_size=2)
