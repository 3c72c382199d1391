"""Process-group topology registry for one-process-per-GPU training on MI355X nodes.

Capability parity with the reference ProcessTopology (``tpc``) singleton
(/root/reference/torchdistpackage/dist/process_topo.py:53-259): ordered
``dist_config`` -> named process groups ('data','pipe','tensor', derived 'model',
plus 'moe_dp'/'moe_ep'), rank/size/neighbor queries, and a ``test_comm`` probe.

Design notes (MI355X-first, not a translation):

- Groups are plain ``torch.distributed`` process groups; with backend "nccl" on
  ROCm these are RCCL communicators over xGMI.  A single 8-GPU MI355X node is
  fully connected (7 p2p links/GPU), so *any* axis ordering gives every group a
  direct-link ring; the ordering convention only decides which ranks share the
  innermost (most bandwidth-hungry) axis.
- The last entry of ``dist_config`` is the innermost axis (adjacent global
  ranks).  Put 'tensor' last: TP all-reduce/all-gather fires per layer and
  benefits most from dense, adjacent communicators.
- Group handles are cached by (mode) name; every rank participates in every
  ``dist.new_group`` call, as required by torch.distributed.
"""

from __future__ import annotations

import datetime
import os
from typing import Dict, List, Optional, Sequence, Tuple

import torch
import torch.distributed as dist

_GROUP_TIMEOUT_S = int(os.environ.get("TDPA_NEW_GROUP_TIMEOUT", "100"))


def _product(xs) -> int:
    out = 1
    for x in xs:
        out *= int(x)
    return out


def gen_axis_groups(world_size: int, axis_size: int, inner_stride: int) -> List[List[int]]:
    """Enumerate the rank lists of one topology axis.

    The global rank space is a row-major multi-index over the ordered axes.  An
    axis with ``inner_stride`` = product of the sizes of all axes *after* it
    groups ranks ``base + i*inner_stride`` for i in [0, axis_size).

    Returns a list of ``world_size // axis_size`` rank lists covering all ranks.
    """
    if world_size % (axis_size * inner_stride) != 0:
        raise ValueError(
            f"world_size={world_size} not divisible by axis_size*stride="
            f"{axis_size}*{inner_stride}"
        )
    groups = []
    span = axis_size * inner_stride
    for block in range(world_size // span):
        for inner in range(inner_stride):
            base = block * span + inner
            groups.append([base + i * inner_stride for i in range(axis_size)])
    return groups


class _Singleton(type):
    _instances: Dict[type, object] = {}

    def __call__(cls, *args, **kwargs):
        if cls not in cls._instances:
            cls._instances[cls] = super().__call__(*args, **kwargs)
        return cls._instances[cls]


class ProcessTopology(metaclass=_Singleton):
    """Named process-group registry.

    Usage::

        tpc.setup_process_groups([('data', 2), ('pipe', 2), ('tensor', 2)])
        tp_group = tpc.get_group('tensor')
        dp_rank = tpc.get_dp_rank()
    """

    def __init__(self):
        self._reset()

    # -- lifecycle -------------------------------------------------------

    def _reset(self):
        self._groups: Dict[str, dist.ProcessGroup] = {}
        self._ranks_in_group: Dict[str, List[int]] = {}
        self._ranks_all: Dict[str, List[List[int]]] = {}
        self._group_rank: Dict[str, int] = {}
        self._group_size: Dict[str, int] = {}
        self._axis_order: List[str] = []
        self._axis_sizes: Dict[str, int] = {}
        self._inited = False

    def destroy(self):
        """Forget all registered groups (for tests that re-init topology)."""
        self._reset()

    # -- construction ----------------------------------------------------

    def setup_process_groups(self, dist_config: Sequence[Tuple[str, int]]):
        """Build named groups from an ordered (name, size) config.

        Order = topology: the last entry is the innermost axis (adjacent global
        ranks); an axis's stride is the product of the sizes after it.  Also
        derives the 'model' axis = all ranks sharing a 'data' coordinate
        (transpose of the data groups), matching the reference behaviour
        (process_topo.py:112-116).
        """
        if not dist.is_initialized():
            raise RuntimeError("torch.distributed must be initialized first")
        world_size = dist.get_world_size()
        sizes = [int(s) for _, s in dist_config]
        names = [n for n, _ in dist_config]
        if len(set(names)) != len(names):
            raise ValueError(f"duplicate axis names in {names}")
        if _product(sizes) != world_size:
            raise ValueError(
                f"product of dist_config sizes {sizes} != world_size {world_size}"
            )
        self._reset()
        self._axis_order = list(names)
        self._axis_sizes = dict(zip(names, sizes))

        for idx, (name, size) in enumerate(zip(names, sizes)):
            inner_stride = _product(sizes[idx + 1:])
            self._build_axis(name, size, inner_stride, world_size)

        # Derived 'model' axis: everything that is not data-parallel.  The
        # model groups are the transpose of the data groups: ranks that share a
        # data-group index.
        if "data" in self._axis_sizes and "model" not in self._axis_sizes:
            dp = self._axis_sizes["data"]
            mp = world_size // dp
            data_groups = gen_axis_groups(
                world_size, dp,
                _product([self._axis_sizes[n] for n in self._axis_order
                          [self._axis_order.index("data") + 1:]]),
            )
            # transpose: model group j = {data_groups[i][j'] ...} -> collect by
            # position within each data group
            model_groups = [[g[i] for g in data_groups] for i in range(dp)]
            # model_groups above has dp lists of len num_data_groups == mp
            self._register_axis("model", [sorted(g) for g in model_groups], mp)
        self._inited = True

    def _build_axis(self, name: str, size: int, inner_stride: int, world_size: int):
        rank_lists = gen_axis_groups(world_size, size, inner_stride)
        self._register_axis(name, rank_lists, size)

    def _register_axis(self, name: str, rank_lists: List[List[int]], size: int):
        rank = dist.get_rank()
        timeout = datetime.timedelta(seconds=_GROUP_TIMEOUT_S)
        for ranks in rank_lists:
            grp = dist.new_group(ranks=ranks, timeout=timeout)
            if rank in ranks:
                self._groups[name] = grp
                self._ranks_in_group[name] = list(ranks)
                self._group_rank[name] = ranks.index(rank)
                self._group_size[name] = len(ranks)
        # Axis present globally even if this rank's group wasn't matched (can't
        # happen — rank lists cover all ranks — but keep the invariant checked).
        assert name in self._groups, f"rank {rank} not covered by axis {name}"
        self._ranks_all[name] = [list(r) for r in rank_lists]
        self._axis_sizes.setdefault(name, size)

    def build_moe_groups(self, moe_dp_size: int, moe_ep_size: int):
        """Split each data group into EP subgroups (contiguous within the data
        group) and MoE-DP subgroups (strided across EP peers).

        Reference behaviour: process_topo.py:118-143.  EP groups hold the
        experts' all-to-all; MoE-DP groups all-reduce replicated expert grads.
        """
        if "data" not in self._groups:
            raise RuntimeError("data axis required before build_moe_groups")
        dp_size = self._group_size["data"]
        if moe_dp_size * moe_ep_size != dp_size:
            raise ValueError(
                f"moe_dp({moe_dp_size}) * moe_ep({moe_ep_size}) != dp({dp_size})"
            )
        world_size = dist.get_world_size()
        rank = dist.get_rank()
        timeout = datetime.timedelta(seconds=_GROUP_TIMEOUT_S)

        # Enumerate all data groups (all ranks must call new_group for all).
        data_stride = _product(
            [self._axis_sizes[n] for n in
             self._axis_order[self._axis_order.index("data") + 1:]]
        )
        all_data_groups = gen_axis_groups(world_size, dp_size, data_stride)

        for mode, rank_sel in (
            ("moe_ep", lambda g: [g[e * moe_ep_size:(e + 1) * moe_ep_size]
                                  for e in range(moe_dp_size)]),
            ("moe_dp", lambda g: [g[i::moe_ep_size] for i in range(moe_ep_size)]),
        ):
            for dg in all_data_groups:
                for ranks in rank_sel(dg):
                    grp = dist.new_group(ranks=ranks, timeout=timeout)
                    if rank in ranks:
                        self._groups[mode] = grp
                        self._ranks_in_group[mode] = list(ranks)
                        self._group_rank[mode] = ranks.index(rank)
                        self._group_size[mode] = len(ranks)

    # -- queries ---------------------------------------------------------

    def is_mode_inited(self, mode: str) -> bool:
        return mode in self._groups

    def get_group(self, mode: str) -> dist.ProcessGroup:
        return self._groups[mode]

    def get_ranks_in_group(self, mode: str) -> List[int]:
        return self._ranks_in_group[mode]

    def get_group_rank(self, mode: str) -> int:
        return self._group_rank[mode]

    def get_group_size(self, mode: str) -> int:
        if mode not in self._group_size:
            return 1
        return self._group_size[mode]

    # convenience accessors, reference-compatible names
    def get_dp_rank(self) -> int:
        return self._group_rank.get("data", 0)

    def get_dp_size(self) -> int:
        return self._group_size.get("data", 1)

    def get_tp_rank(self) -> int:
        return self._group_rank.get("tensor", 0)

    def get_tp_size(self) -> int:
        return self._group_size.get("tensor", 1)

    def get_pp_rank(self) -> int:
        return self._group_rank.get("pipe", 0)

    def get_pp_size(self) -> int:
        return self._group_size.get("pipe", 1)

    def get_mp_rank(self) -> int:
        return self._group_rank.get("model", 0)

    def get_mp_size(self) -> int:
        return self._group_size.get("model", 1)

    def is_first_in_pipeline_group(self) -> bool:
        return self.get_pp_rank() == 0

    def is_last_in_pipeline_group(self) -> bool:
        return self.get_pp_rank() == self.get_pp_size() - 1

    def is_first_in_data_group(self) -> bool:
        return self.get_dp_rank() == 0

    def is_first_in_tensor_group(self) -> bool:
        return self.get_tp_rank() == 0

    def get_prev_global_rank(self, mode: str = "pipe") -> int:
        """Global rank of the ring-predecessor within this rank's mode group."""
        ranks = self._ranks_in_group[mode]
        i = self._group_rank[mode]
        return ranks[(i - 1) % len(ranks)]

    def get_next_global_rank(self, mode: str = "pipe") -> int:
        ranks = self._ranks_in_group[mode]
        i = self._group_rank[mode]
        return ranks[(i + 1) % len(ranks)]

    def all_ranks(self, mode: Optional[str] = None):
        """No mode: all global ranks.  With a mode: every group's rank list
        for that axis (reference all_ranks, process_topo.py:242-246)."""
        if mode is None:
            return list(range(dist.get_world_size()))
        if mode not in self._ranks_all:
            raise RuntimeError(f"{mode} is not initialized")
        return self._ranks_all[mode]

    def all_dp_ranks(self) -> List[List[int]]:
        return self.all_ranks("data")

    def is_first_in_group(self, mode: str) -> bool:
        return self._group_rank.get(mode, 0) == 0

    def is_last_in_group(self, mode: str) -> bool:
        return self._group_rank.get(mode, 0) == self._group_size.get(mode, 1) - 1

    def is_last_in_data_group(self) -> bool:
        return self.is_last_in_group("data")

    def is_last_in_tensor_group(self) -> bool:
        return self.is_last_in_group("tensor")

    def is_first_in_model_group(self) -> bool:
        return self.is_first_in_group("model")

    def is_last_in_model_group(self) -> bool:
        return self.is_last_in_group("model")

    def is_first_group(self, mode: str) -> bool:
        """True if this rank's mode-group IS the first group of that axis
        (reference process_topo.py:248-259)."""
        if mode not in self._ranks_all:
            raise RuntimeError(f"{mode} is not initialized")
        return self._ranks_in_group[mode] == self._ranks_all[mode][0]

    @property
    def inited(self) -> bool:
        return self._inited


tpc = ProcessTopology()
# reference-compatible alias (process_topo.py:262)
torch_parallel_context = tpc


def is_using_pp() -> bool:
    return tpc.is_mode_inited("pipe") and tpc.get_pp_size() > 1


def test_comm(verbose: bool = False):
    """Probe every initialized group with all-reduce, broadcast, all-gather
    and ring p2p (send/recv to group neighbours).

    Reference parity: process_topo.py:267-316 (which also probes p2p and
    all-gather — round 1 covered only all-reduce + broadcast).  Uses a
    (100, 128) tensor on the current device.
    """
    if not dist.is_initialized():
        raise RuntimeError("torch.distributed not initialized")
    dev = torch.device("cuda", torch.cuda.current_device()) \
        if torch.cuda.is_available() else torch.device("cpu")
    rank = dist.get_rank()
    for mode, group in tpc._groups.items():
        ranks = tpc.get_ranks_in_group(mode)
        t = torch.full((100, 128), float(rank), device=dev)
        dist.all_reduce(t, group=group)
        expected = float(sum(ranks))
        assert torch.allclose(t, torch.full_like(t, expected)), \
            f"all_reduce mismatch in group {mode}"
        b = torch.full((8,), float(ranks[0]), device=dev)
        dist.broadcast(b, src=ranks[0], group=group)
        assert torch.allclose(b, torch.full_like(b, float(ranks[0]))), \
            f"broadcast mismatch in group {mode}"
        # all-gather: every member's rank shows up in its slot
        g = torch.full((4,), float(rank), device=dev)
        out = [torch.empty_like(g) for _ in ranks]
        dist.all_gather(out, g, group=group)
        for slot, r in zip(out, ranks):
            assert torch.allclose(slot, torch.full_like(slot, float(r))), \
                f"all_gather mismatch in group {mode}"
        # ring p2p: send to next, receive from prev (skip trivial groups)
        if len(ranks) > 1:
            idx = ranks.index(rank)
            nxt = ranks[(idx + 1) % len(ranks)]
            prv = ranks[(idx - 1) % len(ranks)]
            s = torch.full((16,), float(rank), device=dev)
            r = torch.empty_like(s)
            ops = [dist.P2POp(dist.isend, s, nxt),
                   dist.P2POp(dist.irecv, r, prv)]
            for w in dist.batch_isend_irecv(ops):
                w.wait()
            assert torch.allclose(r, torch.full_like(r, float(prv))), \
                f"p2p ring mismatch in group {mode}"
        if verbose and rank == 0:
            print(f"[test_comm] group '{mode}' ok (ranks={ranks})")
    dist.barrier()
