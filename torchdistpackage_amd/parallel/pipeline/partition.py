"""Model flattening + stage partitioning for pipeline parallelism.

Capability parity with the reference partitioner
(/root/reference/torchdistpackage/parallel/pipeline_parallel/
pipeline_helper.py): recursive ``flatten_sequence``, ``flatten_model`` by
layer-name exec list (lambdas wrapped as CallableModule), ``partition_uniform``
equal-count slicing, ``partition_balanced`` param-count balancing, and the
``flat_and_partition`` dispatcher.
"""

from __future__ import annotations

from typing import Callable, List, Optional, Sequence, Union

import torch.nn as nn

from ...dist.topo import tpc


class CallableModule(nn.Module):
    """Wraps a bare callable (lambda/function) as an nn.Module so it can live
    in a stage nn.Sequential (reference pipeline_helper.py:131-176)."""

    def __init__(self, fn: Callable):
        super().__init__()
        self.fn = fn

    def forward(self, *args, **kwargs):
        return self.fn(*args, **kwargs)


def flatten_sequence(model: Union[nn.Module, Sequence]) -> List[nn.Module]:
    """Recursively flatten nn.Sequential / ModuleList nesting into a flat
    layer list (reference pipeline_helper.py:114-128)."""
    if isinstance(model, (nn.Sequential, nn.ModuleList)):
        out: List[nn.Module] = []
        for m in model:
            out.extend(flatten_sequence(m))
        return out
    if isinstance(model, (list, tuple)):
        out = []
        for m in model:
            out.extend(flatten_sequence(m))
        return out
    return [model]


def flatten_model(model: nn.Module,
                  exec_order: Sequence[Union[str, Callable]]) -> List[nn.Module]:
    """Flatten by an explicit execution list of attribute names / callables.

    ``exec_order`` entries: a dotted attribute path into ``model`` (resolved
    to that submodule; nn.Sequential/ModuleList entries are inlined), or a
    bare callable (wrapped as CallableModule).
    """
    layers: List[nn.Module] = []
    for entry in exec_order:
        if callable(entry) and not isinstance(entry, nn.Module):
            layers.append(CallableModule(entry))
            continue
        obj = model
        for attr in entry.split("."):
            obj = obj[int(attr)] if attr.isdigit() else getattr(obj, attr)
        layers.extend(flatten_sequence(obj))
    return layers


def partition_uniform(num_layers: int, num_stages: int) -> List[List[int]]:
    """Equal-count contiguous slices [start, end) per stage
    (reference pipeline_helper.py:6-17)."""
    base = num_layers // num_stages
    rem = num_layers % num_stages
    parts = []
    start = 0
    for s in range(num_stages):
        n = base + (1 if s < rem else 0)
        parts.append([start, start + n])
        start += n
    return parts


def partition_balanced(layers: Sequence[nn.Module],
                       num_stages: int) -> List[List[int]]:
    """Param-count-balanced contiguous partition (reference
    pipeline_helper.py:20-111): minimize the max per-stage weight via binary
    search over the bottleneck with a greedy feasibility check."""
    weights = [max(sum(p.numel() for p in l.parameters()), 1) for l in layers]
    n = len(weights)
    assert n >= num_stages, f"{n} layers < {num_stages} stages"

    def feasible(cap: int) -> Optional[List[List[int]]]:
        parts = []
        start = 0
        acc = 0
        for i, w in enumerate(weights):
            if w > cap:
                return None
            if acc + w > cap:
                parts.append([start, i])
                start = i
                acc = 0
            acc += w
        parts.append([start, n])
        if len(parts) > num_stages:
            return None
        while len(parts) < num_stages:  # pad empty stages at the end
            parts.append([n, n])
        return parts

    lo, hi = max(weights), sum(weights)
    best = feasible(hi)
    while lo <= hi:
        mid = (lo + hi) // 2
        f = feasible(mid)
        if f is not None:
            best = f
            hi = mid - 1
        else:
            lo = mid + 1
    return best


def partition_by_time(layers: Sequence[nn.Module], num_stages: int,
                      sample_input, warmup: int = 2,
                      iters: int = 5) -> List[List[int]]:
    """Contiguous partition balanced by MEASURED per-layer forward time.

    Capability parity with the reference's fx-based auto-split
    (/root/reference/explore/fx/fx_graph_split.py:123-172), without fx: the
    layer list is executed sequentially on ``sample_input`` and timed (device
    sync per layer), then the same bottleneck binary search as
    partition_balanced slices by time.
    """
    import time as _time
    import torch as _torch
    x = sample_input
    times = [0.0] * len(layers)
    with _torch.no_grad():
        for it in range(warmup + iters):
            xi = x
            for li, layer in enumerate(layers):
                if _torch.cuda.is_available():
                    _torch.cuda.synchronize()
                t0 = _time.perf_counter()
                xi = layer(xi)
                if _torch.cuda.is_available():
                    _torch.cuda.synchronize()
                if it >= warmup:
                    times[li] += _time.perf_counter() - t0
    # reuse the bottleneck search with time weights (scaled to ints)
    scale = 1e7
    weights = [max(int(t * scale), 1) for t in times]
    n = len(weights)

    def feasible(cap):
        parts, start, acc = [], 0, 0
        for i, w in enumerate(weights):
            if w > cap:
                return None
            if acc + w > cap:
                parts.append([start, i])
                start, acc = i, 0
            acc += w
        parts.append([start, n])
        if len(parts) > num_stages:
            return None
        while len(parts) < num_stages:
            parts.append([n, n])
        return parts

    lo, hi = max(weights), sum(weights)
    best = feasible(hi)
    while lo <= hi:
        mid = (lo + hi) // 2
        f = feasible(mid)
        if f is not None:
            best, hi = f, mid - 1
        else:
            lo = mid + 1
    return best


def flat_and_partition(model: nn.Module, num_stages: Optional[int] = None,
                       method: str = "uniform",
                       exec_order: Optional[Sequence] = None) -> nn.Sequential:
    """Flatten + slice out THIS pp rank's stage as an nn.Sequential
    (reference pipeline_helper.py:179-183)."""
    if num_stages is None:
        num_stages = tpc.get_pp_size()
    if exec_order is not None:
        layers = flatten_model(model, exec_order)
    elif hasattr(model, "to_stage_layers"):
        layers = model.to_stage_layers()
    else:
        layers = flatten_sequence(model)
    if method == "balanced":
        parts = partition_balanced(layers, num_stages)
    else:
        parts = partition_uniform(len(layers), num_stages)
    start, end = parts[tpc.get_pp_rank()]
    return nn.Sequential(*layers[start:end])
