"""Context-parallel attention tests (gloo, CPU): Ulysses and ring attention
vs the full-sequence oracle, forward AND backward."""

import math

import torch

from tests.dist_helpers import run_distributed


def _full_oracle(q, k, v, causal):
    scale = 1.0 / math.sqrt(q.shape[-1])
    s = torch.matmul(q.float(), k.float().transpose(-1, -2)) * scale
    if causal:
        S = q.shape[-2]
        mask = torch.ones(S, S, dtype=torch.bool).tril_()
        s = s.masked_fill(~mask, float("-inf"))
    return torch.matmul(torch.softmax(s, -1), v.float())


def _cp_case(rank, world_size, kind="ulysses", causal=True):
    import torch.distributed as dist
    from torchdistpackage_amd.dist.topo import tpc
    from torchdistpackage_amd.parallel.context import (ulysses_attention,
                                                       ring_attention)

    tpc.setup_process_groups([("context", world_size)])
    group = tpc.get_group("context")
    B, H, S, D = 2, 4, 64, 32
    S_loc = S // world_size
    torch.manual_seed(0)
    q = torch.randn(B, H, S, D)
    k = torch.randn(B, H, S, D)
    v = torch.randn(B, H, S, D)

    def shard(t):
        return t[:, :, rank * S_loc:(rank + 1) * S_loc].clone() \
            .requires_grad_(True)

    qs, ks, vs = shard(q), shard(k), shard(v)
    fn = ulysses_attention if kind == "ulysses" else ring_attention
    o = fn(qs, ks, vs, causal=causal, group=group)

    qf = q.clone().requires_grad_(True)
    kf = k.clone().requires_grad_(True)
    vf = v.clone().requires_grad_(True)
    ref = _full_oracle(qf, kf, vf, causal)
    ref_shard = ref[:, :, rank * S_loc:(rank + 1) * S_loc]
    assert torch.allclose(o.float(), ref_shard, atol=2e-4), \
        f"{kind} fwd mismatch {(o.float() - ref_shard).abs().max().item()}"

    torch.manual_seed(7)
    g_full = torch.randn(B, H, S, D)
    g = g_full[:, :, rank * S_loc:(rank + 1) * S_loc]
    o.backward(g)
    ref.backward(g_full.float())

    for got, full_grad, name in ((qs.grad, qf.grad, "dq"),
                                 (ks.grad, kf.grad, "dk"),
                                 (vs.grad, vf.grad, "dv")):
        want = full_grad[:, :, rank * S_loc:(rank + 1) * S_loc]
        err = (got.float() - want).abs().max().item()
        assert err < 5e-4, f"{kind} {name} err {err}"
    return True


def test_ulysses_causal():
    run_distributed(_cp_case, world_size=2, kwargs={"kind": "ulysses"})


def test_ulysses_bidirectional():
    run_distributed(_cp_case, world_size=2,
                    kwargs={"kind": "ulysses", "causal": False})


def test_ring_causal():
    run_distributed(_cp_case, world_size=2, kwargs={"kind": "ring"})


def test_ring_bidirectional():
    run_distributed(_cp_case, world_size=2,
                    kwargs={"kind": "ring", "causal": False})


def test_ring_causal_cp4():
    run_distributed(_cp_case, world_size=4, kwargs={"kind": "ring"})
