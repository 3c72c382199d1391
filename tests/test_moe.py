"""MoE layer tests: EP dispatch vs the dense oracle (gloo world_size=2, CPU).

Oracle method: the same experts evaluated WITHOUT any parallelism/dispatch
(every token sent to its experts locally) must produce identical output.
"""

import copy

import torch
import torch.nn as nn

from tests.dist_helpers import run_distributed
from torchdistpackage_amd.moe import ExpertParallelMoE, TopKRouter


def _dense_oracle(moe_layers, router_out, x):
    """Evaluate routing decisions against a full local expert list."""
    topk_idx, topk_gate, _ = router_out
    out = torch.zeros_like(x)
    for k in range(topk_idx.shape[1]):
        for e, expert in enumerate(moe_layers):
            mask = topk_idx[:, k] == e
            if mask.any():
                y = expert(x[mask])
                out[mask] += topk_gate[mask, k].unsqueeze(-1).to(y.dtype) * y
    return out


def test_router_topk():
    torch.manual_seed(0)
    r = TopKRouter(16, 8, top_k=2)
    x = torch.randn(32, 16)
    idx, gate, aux = r(x)
    assert idx.shape == (32, 2) and gate.shape == (32, 2)
    assert torch.allclose(gate.sum(-1), torch.ones(32), atol=1e-5)
    assert aux.item() > 0


def test_moe_single_process():
    """ep_size=1: dispatch must equal the dense oracle exactly."""
    torch.manual_seed(1)
    dim = 32
    moe = ExpertParallelMoE(dim, num_experts=4, top_k=2, hidden_mult=2)
    x = torch.randn(6, 3, dim)
    out = moe(x)
    xt = x.reshape(-1, dim)
    router_out = moe.router(xt)  # same weights -> same decisions
    ref = _dense_oracle(list(moe.experts), router_out, xt)
    assert torch.allclose(out.reshape(-1, dim), ref, atol=1e-5), \
        (out.reshape(-1, dim) - ref).abs().max().item()


def _moe_ep2(rank, world_size):
    import torch.distributed as dist
    from torchdistpackage_amd.dist.topo import tpc
    from torchdistpackage_amd.moe import ExpertParallelMoE

    tpc.setup_process_groups([("data", world_size)])
    tpc.build_moe_groups(moe_dp_size=1, moe_ep_size=world_size)

    dim, E = 32, 4
    torch.manual_seed(0)  # same router everywhere
    moe = ExpertParallelMoE(dim, num_experts=E, top_k=2, hidden_mult=2,
                            ep_group=tpc.get_group("moe_ep"))
    assert moe.num_local == E // world_size

    # build the full expert list (as a dense oracle) deterministically:
    # rank r holds local experts [r*num_local, (r+1)*num_local); re-seed and
    # construct all E experts the same way each rank constructed its own.
    from torchdistpackage_amd.moe.layer import Expert
    torch.manual_seed(100)
    all_experts = [Expert(dim, 2) for _ in range(E)]
    # overwrite this rank's shard with the canonical weights
    with torch.no_grad():
        for i, exp in enumerate(moe.experts):
            ge = all_experts[moe.ep_rank * moe.num_local + i]
            exp.fc1.weight.copy_(ge.fc1.weight)
            exp.fc1.bias.copy_(ge.fc1.bias)
            exp.fc2.weight.copy_(ge.fc2.weight)
            exp.fc2.bias.copy_(ge.fc2.bias)

    # identical input on all EP ranks (moe_ep splits the dp group; within an
    # EP group each rank feeds its own tokens — here same tokens to compare)
    torch.manual_seed(5 + rank)   # DIFFERENT tokens per rank (realistic)
    x = torch.randn(4, 2, dim)
    out = moe(x)

    xt = x.reshape(-1, dim)
    router_out = moe.router(xt)
    ref = _dense_oracle(all_experts, router_out, xt)
    assert torch.allclose(out.reshape(-1, dim), ref, atol=1e-5), \
        (out.reshape(-1, dim) - ref).abs().max().item()
    return True


def test_moe_ep2_dispatch():
    run_distributed(_moe_ep2, world_size=2)


def _moe_backward(rank, world_size):
    """Gradients flow through the all-to-all: grads of expert weights match
    the dense oracle."""
    import torch.distributed as dist
    from torchdistpackage_amd.dist.topo import tpc
    from torchdistpackage_amd.moe import ExpertParallelMoE
    from torchdistpackage_amd.moe.layer import Expert

    tpc.setup_process_groups([("data", world_size)])
    tpc.build_moe_groups(moe_dp_size=1, moe_ep_size=world_size)
    dim, E = 16, 2
    torch.manual_seed(0)
    moe = ExpertParallelMoE(dim, num_experts=E, top_k=1, hidden_mult=2,
                            ep_group=tpc.get_group("moe_ep"))
    torch.manual_seed(100)
    all_experts = [Expert(dim, 2) for _ in range(E)]
    with torch.no_grad():
        for i, exp in enumerate(moe.experts):
            ge = all_experts[moe.ep_rank * moe.num_local + i]
            exp.load_state_dict(ge.state_dict())

    torch.manual_seed(7 + rank)
    x = torch.randn(6, dim, requires_grad=True)
    out = moe(x)
    out.pow(2).sum().backward()

    # oracle on the union of both ranks' tokens
    xs = []
    for r in range(world_size):
        torch.manual_seed(7 + r)
        xs.append(torch.randn(6, dim))
    x_all = torch.cat(xs).requires_grad_(True)
    router_out = moe.router(x_all)
    ref = _dense_oracle(all_experts, router_out, x_all)
    ref.pow(2).sum().backward()

    # this rank's local expert grads must equal the oracle's (tokens from
    # BOTH ranks hit it)
    for i, exp in enumerate(moe.experts):
        ge = all_experts[moe.ep_rank * moe.num_local + i]
        assert torch.allclose(exp.fc1.weight.grad, ge.fc1.weight.grad,
                              atol=1e-4), \
            (exp.fc1.weight.grad - ge.fc1.weight.grad).abs().max().item()
    # dx for this rank's tokens
    my = x_all.grad[rank * 6:(rank + 1) * 6]
    assert torch.allclose(x.grad, my, atol=1e-4)
    return True


def test_moe_backward_ep2():
    run_distributed(_moe_backward, world_size=2)


def test_moe_model_forward():
    from torchdistpackage_amd.models.moe_model import MoEConfig, MoEModel
    cfg = MoEConfig(vocab_size=128, n_layer=2, n_head=2, dim=32, max_seq=16,
                    num_experts=4, top_k=2, hidden_mult=2)
    torch.manual_seed(0)
    m = MoEModel(cfg)
    x = torch.randint(0, 128, (2, 16))
    out = m(x, labels=x)
    assert out["loss"].item() > 0
    out["loss"].backward()
    # expert params got grads and are tagged
    n_expert = sum(1 for p in m.expert_parameters())
    assert n_expert == 2 * 4 * 4  # layers * experts * (2 linears w+b)
    for p in m.expert_parameters():
        assert p.grad is not None


def _moe_full_composition(rank, world_size):
    """dp4 = moe_dp2 x moe_ep2 full training composition: dense params sync
    over 'data' (NaiveDdp), expert params over 'moe_dp' (MoEDP); two steps
    run and the replicated expert shards stay identical across moe_dp peers."""
    import torch.distributed as dist
    from torchdistpackage_amd.dist.topo import tpc
    from torchdistpackage_amd.ddp import NaiveDdp, create_moe_dp_hooks, \
        moe_dp_iter_step
    from torchdistpackage_amd.models.moe_model import MoEConfig, MoEModel

    tpc.setup_process_groups([("data", world_size)])
    tpc.build_moe_groups(moe_dp_size=2, moe_ep_size=2)
    cfg = MoEConfig(vocab_size=64, n_layer=1, n_head=2, dim=16, max_seq=8,
                    num_experts=4, top_k=1, hidden_mult=2)
    torch.manual_seed(0)  # same init; broadcasts enforce it anyway
    model = MoEModel(cfg)
    ddp = NaiveDdp(model)
    create_moe_dp_hooks(list(model.expert_parameters()))
    opt = torch.optim.SGD(model.parameters(), lr=0.1)

    for it in range(2):
        torch.manual_seed(it * 10 + rank)
        x = torch.randint(0, 64, (2, 8))
        out = ddp(x, labels=x)
        out["loss"].backward()
        ddp.reduce_gradients()
        moe_dp_iter_step()
        opt.step()
        opt.zero_grad()

    # expert params must be identical across the moe_dp group
    grp = tpc.get_group("moe_dp")
    for p in model.expert_parameters():
        mine = p.detach().clone()
        peers = [torch.zeros_like(mine) for _ in range(2)]
        dist.all_gather(peers, mine, group=grp)
        assert torch.allclose(peers[0], peers[1], atol=1e-6)
    # dense params identical across the FULL data group
    for p in model.non_expert_parameters():
        mine = p.detach().clone()
        peers = [torch.zeros_like(mine) for _ in range(world_size)]
        dist.all_gather(peers, mine)
        for pr in peers[1:]:
            assert torch.allclose(peers[0], pr, atol=1e-6)
    return True


def test_moe_full_composition_world4():
    run_distributed(_moe_full_composition, world_size=4)


def _bench_order_regression(rank, world_size):
    """Regression: bench.py once built the MoE model BEFORE build_moe_groups,
    so layers captured ep_size=1 with group=None and the dispatch emulation
    treated None as the world group (IndexError).  Groups-first must work and
    the single-EP capture must be harmless."""
    import torch
    from torchdistpackage_amd.dist.topo import tpc
    from torchdistpackage_amd.models.moe_model import MoEConfig, MoEModel

    tpc.setup_process_groups([("data", world_size)])
    tpc.build_moe_groups(moe_dp_size=1, moe_ep_size=world_size)
    cfg = MoEConfig(vocab_size=64, n_layer=1, n_head=2, dim=32, max_seq=16,
                    num_experts=world_size * 2, top_k=2, hidden_mult=2)
    m = MoEModel(cfg)
    x = torch.randint(0, 64, (2, 16))
    out = m(x, labels=x)
    out["loss"].backward()
    assert torch.isfinite(out["loss"])

    return True


def test_bench_order_regression():
    run_distributed(_bench_order_regression, world_size=2)


def _moe_ep1_in_multirank_world(rank, world_size):
    """ADVICE r01 regression: ExpertParallelMoE with ep_group=None (no moe
    groups built) inside a multi-rank job must NOT issue the combine
    all-to-all over WORLD (it crashed / mis-split before the guard)."""
    from torchdistpackage_amd.moe.layer import ExpertParallelMoE

    torch.manual_seed(5)
    moe = ExpertParallelMoE(dim=16, num_experts=4, top_k=2, ep_group=None)
    assert moe.ep_size == 1
    x = torch.randn(3, 7, 16)
    y = moe(x)
    assert y.shape == x.shape
    assert torch.isfinite(y).all()
    y.sum().backward()
    return True


def test_moe_ep1_guard_world2():
    run_distributed(_moe_ep1_in_multirank_world, world_size=2)


def test_batched_experts_parity():
    """BatchedExperts (two baddbmm over all local experts, clamp-gather
    padding) vs the per-expert ModuleList path, identical weights."""
    from torchdistpackage_amd.moe.layer import ExpertParallelMoE
    torch.manual_seed(13)
    dim = 32
    a = ExpertParallelMoE(dim, num_experts=4, top_k=2, hidden_mult=2,
                          ep_group=None, batched=False)
    b = ExpertParallelMoE(dim, num_experts=4, top_k=2, hidden_mult=2,
                          ep_group=None, batched=True)
    b.load_state_dict({k: v for k, v in a.state_dict().items()
                       if k.startswith("router")}, strict=False)
    b.experts_b.load_from_experts(a.experts)
    with torch.no_grad():
        b.router.gate.weight.copy_(a.router.gate.weight)

    x = torch.randn(6, 5, dim, requires_grad=True)
    x2 = x.detach().clone().requires_grad_(True)
    ya = a(x)
    yb = b(x2)
    assert torch.allclose(ya, yb, atol=1e-5), (ya - yb).abs().max()
    g = torch.randn_like(ya)
    ya.backward(g)
    yb.backward(g)
    assert torch.allclose(x.grad, x2.grad, atol=1e-5)
    # expert grads match: fc1 of expert e <-> w1[e]
    for e, ex in enumerate(a.experts):
        assert torch.allclose(b.experts_b.w1.grad[e], ex.fc1.weight.grad,
                              atol=1e-5), e
        assert torch.allclose(b.experts_b.b2.grad[e], ex.fc2.bias.grad,
                              atol=1e-5), e


def test_batched_experts_empty_expert():
    """Empty experts (routing collapse) must not index past the token
    buffer — the bench crashed on exactly this before the clamp."""
    from torchdistpackage_amd.moe.layer import BatchedExperts
    torch.manual_seed(17)
    be = BatchedExperts(num_local=4, dim=16, hidden_mult=2)
    grouped = torch.randn(10, 16)
    # experts 2 and 3 (the LAST) empty
    cnt = torch.tensor([4, 6, 0, 0])
    y = be(grouped, cnt, 6)
    assert y.shape == (10, 16)
    assert torch.isfinite(y).all()
    # all-empty tail + zero tokens entirely
    y2 = be(grouped[:0], torch.zeros(4, dtype=torch.long), 0)
    assert y2.shape == (0, 16)
