"""NaiveDdp differential tests vs torch DDP (gloo, world_size=2, CPU).

Mirrors the reference's examples/test_ddp.py method: same deep-copied model,
identical inputs per rank, per-iter assert grads/params allclose between
NaiveDdp and torch.nn.parallel.DistributedDataParallel.
This is BASELINE.json config 1 ("NaiveDdp on 2-layer MLP, gloo world_size=2").
"""

import copy

import torch
import torch.nn as nn

from tests.dist_helpers import run_distributed


def _make_mlp(seed=0):
    torch.manual_seed(seed)
    return nn.Sequential(nn.Linear(64, 128), nn.ReLU(), nn.Linear(128, 32))


def _ddp_vs_torch(rank, world_size, sync=False, bucket_cap_mb=50.0):
    import torch.distributed as dist
    from torchdistpackage_amd.ddp import NaiveDdp

    torch.manual_seed(1234)  # same init everywhere; broadcast also covers it
    model_a = _make_mlp(seed=7)
    model_b = copy.deepcopy(model_a)

    mine = NaiveDdp(model_a, sync=sync, bucket_cap_mb=bucket_cap_mb)
    ref = nn.parallel.DistributedDataParallel(model_b)
    opt_a = torch.optim.AdamW(model_a.parameters(), lr=1e-3)
    opt_b = torch.optim.AdamW(model_b.parameters(), lr=1e-3)

    for it in range(5):
        # different data per rank (the whole point of DP)
        torch.manual_seed(100 + 10 * it + rank)
        x = torch.randn(8, 64)

        out_a = mine(x)
        out_b = ref(x)
        assert torch.allclose(out_a, out_b, atol=1e-6), f"iter {it} fwd"

        out_a.pow(2).mean().backward()
        out_b.pow(2).mean().backward()
        mine.reduce_gradients()

        for (na, pa), (nb, pb) in zip(model_a.named_parameters(),
                                      model_b.named_parameters()):
            assert pa.grad is not None
            assert torch.allclose(pa.grad, pb.grad, atol=1e-5), \
                f"iter {it} grad mismatch {na}"

        opt_a.step()
        opt_b.step()
        opt_a.zero_grad()
        opt_b.zero_grad()

        for (na, pa), (nb, pb) in zip(model_a.named_parameters(),
                                      model_b.named_parameters()):
            assert torch.allclose(pa, pb, atol=1e-5), \
                f"iter {it} param mismatch {na}"
    return True


def test_naive_ddp_bucketed():
    run_distributed(_ddp_vs_torch, world_size=2)


def test_naive_ddp_sync():
    run_distributed(_ddp_vs_torch, world_size=2, kwargs={"sync": True})


def test_naive_ddp_tiny_buckets():
    # force multiple buckets (each param its own bucket at ~0 cap)
    run_distributed(_ddp_vs_torch, world_size=2,
                    kwargs={"bucket_cap_mb": 0.001})


def _grad_acc(rank, world_size):
    """num_grad_acc_iter=2: reduce fires only every 2nd backward; the result
    equals averaging the summed 2-micro-batch grads across ranks."""
    from torchdistpackage_amd.ddp import NaiveDdp

    model = _make_mlp(seed=3)
    ref = copy.deepcopy(model)
    mine = NaiveDdp(model, num_grad_acc_iter=2)

    # two micro-batches per rank, all different
    xs = []
    for mb in range(2):
        torch.manual_seed(1000 + rank * 10 + mb)
        xs.append(torch.randn(4, 64))

    for x in xs:
        mine(x).pow(2).mean().backward()
    mine.reduce_gradients()

    # dense reference: accumulate grads of ALL ranks' micro-batches / world
    for r in range(world_size):
        for mb in range(2):
            torch.manual_seed(1000 + r * 10 + mb)
            x = torch.randn(4, 64)
            ref(x).pow(2).mean().backward()
    for (na, pa), (nb, pb) in zip(model.named_parameters(),
                                  ref.named_parameters()):
        expected = pb.grad / world_size
        assert torch.allclose(pa.grad, expected, atol=1e-5), f"grad {na}"
    return True


def test_naive_ddp_grad_accumulation():
    run_distributed(_grad_acc, world_size=2)


def _stale_partial_bucket(rank, world_size):
    """VERDICT r01 weak #7 regression: a partially-filled bucket flush must
    not reduce (and write back) STALE grads for params whose hooks did not
    fire this iteration."""
    from torchdistpackage_amd.ddp import NaiveDdp

    torch.manual_seed(3)

    class TwoPath(nn.Module):
        def __init__(self):
            super().__init__()
            self.a = nn.Linear(8, 8, bias=False)
            self.b = nn.Linear(8, 8, bias=False)

        def forward(self, x, use_b=True):
            y = self.a(x)
            if use_b:
                y = y + self.b(x)
            return y

    model = NaiveDdp(TwoPath())  # one bucket holds both params

    x = torch.randn(4, 8)
    # iter 1: both params used
    model(x, use_b=True).sum().backward()
    model.reduce_gradients()
    stale_b = model.module.b.weight.grad.clone()
    assert stale_b.abs().sum() > 0
    model.zero_grad(set_to_none=False)  # grads stay allocated (all-zero)

    # iter 2: only 'a' used -> partial bucket flush
    model(x, use_b=False).sum().backward()
    model.reduce_gradients()
    gb = model.module.b.weight.grad
    assert torch.all(gb == 0), \
        f"unused param received stale grad (max {gb.abs().max().item()})"
    ga = model.module.a.weight.grad
    assert ga.abs().sum() > 0
    return True


def test_stale_partial_bucket_flush():
    run_distributed(_stale_partial_bucket, world_size=2)
