"""Property-based tests (hypothesis) for the pure combinatorial helpers:
topology group enumeration, pipeline partitioners, EMA/ZeRO param
partitioning, and the grad-bucket offset table."""

import torch
import torch.nn as nn
from hypothesis import given, settings, strategies as st


# ---------------------------------------------------------------- topology

@st.composite
def _axis_cfg(draw):
    axis = draw(st.integers(1, 8))
    stride = draw(st.integers(1, 8))
    mult = draw(st.integers(1, 6))
    return axis * stride * mult, axis, stride


@given(_axis_cfg())
@settings(max_examples=200, deadline=None)
def test_gen_axis_groups_partitions_world(cfg):
    from torchdistpackage_amd.dist.topo import gen_axis_groups
    world, axis, stride = cfg
    groups = gen_axis_groups(world, axis, stride)
    assert len(groups) == world // axis
    flat = sorted(r for g in groups for r in g)
    assert flat == list(range(world))          # exact cover, no dupes
    for g in groups:
        assert len(g) == axis
        if len(g) > 1:                          # constant stride inside
            deltas = {b - a for a, b in zip(g, g[1:])}
            assert deltas == {stride}


@given(st.integers(1, 64), st.integers(1, 16))
@settings(max_examples=200, deadline=None)
def test_partition_uniform_properties(layers, stages):
    from torchdistpackage_amd.parallel.pipeline import partition_uniform
    parts = partition_uniform(layers, stages)
    assert len(parts) == stages
    assert parts[0][0] == 0 and parts[-1][1] == layers
    for (a, b), (c, d) in zip(parts, parts[1:]):
        assert b == c and a <= b and c <= d     # contiguous, ordered
    sizes = [b - a for a, b in parts]
    assert max(sizes) - min(sizes) <= 1         # equal-count


@given(st.lists(st.integers(1, 512), min_size=1, max_size=24),
       st.integers(1, 8))
@settings(max_examples=100, deadline=None)
def test_partition_balanced_covers_and_bounds(widths, stages):
    from torchdistpackage_amd.parallel.pipeline import partition_balanced
    if len(widths) < stages:
        return
    layers = [nn.Linear(w, 1) for w in widths]
    parts = partition_balanced(layers, stages)
    assert len(parts) == stages
    assert parts[0][0] == 0
    ends = [b for _, b in parts]
    assert max(ends) == len(widths)
    for (a, b), (c, d) in zip(parts, parts[1:]):
        assert b == c                            # contiguous cover
    # bottleneck optimality lower bound: max stage weight >= total/stages
    w = [sum(p.numel() for p in l.parameters()) for l in layers]
    loads = [sum(w[a:b]) for a, b in parts]
    assert max(loads) >= sum(w) / stages - 1e-9


@given(st.lists(st.integers(1, 4096), min_size=1, max_size=40),
       st.integers(1, 8))
@settings(max_examples=100, deadline=None)
def test_partition_by_numel_balance(numels, parts_n):
    from torchdistpackage_amd.dist.sharded_ema import partition_by_numel
    params = [torch.empty(n) for n in numels]
    parts = partition_by_numel(params, parts_n)
    assert len(parts) == parts_n
    flat = sorted(i for p in parts for i in p)
    assert flat == list(range(len(params)))
    loads = [sum(numels[i] for i in p) for p in parts]
    # greedy largest-first: max load <= ideal + largest item
    assert max(loads) <= sum(numels) / parts_n + max(numels)


@given(st.lists(st.integers(1, 300), min_size=1, max_size=12))
@settings(max_examples=100, deadline=None)
def test_grad_bucket_offsets_aligned_disjoint(numels):
    from torchdistpackage_amd.ddp.naive_ddp import GradBucket
    params = [nn.Parameter(torch.zeros(n)) for n in numels]
    b = GradBucket(params, torch.float32, torch.device("cpu"))
    offs = [b.offsets[i] for i in range(len(params))]
    for o in offs:
        assert o * 4 % 512 == 0                 # 512 B alignment
    for i in range(len(params)):
        lo = offs[i]
        hi = lo + numels[i]
        for j in range(i + 1, len(params)):
            assert offs[j] >= hi                # disjoint, ordered
    assert b.data.numel() >= offs[-1] + numels[-1]
