"""Data-parallel-sharded EMA of model parameters.

Capability parity with the reference ShardedEMA
(/root/reference/torchdistpackage/dist/sharded_ema.py + utils.py:35-65): each
DP rank owns a contiguous greedy-balanced subset of params and EMA-updates only
its shard; a full state dict is reassembled on demand.

MI355X-first changes vs the reference:
- The shard is kept as ONE flat fp32 buffer per rank (not per-param clones) so
  the update is a single fused kernel pass over contiguous memory (HBM3E
  bandwidth-bound: one read of shard + one read of params + one write), using
  the in-tree HIP ``ema_update_`` op when the extension is loaded, falling back
  to torch ``lerp_`` otherwise.
- ``state_dict_cpu`` reassembles with one ``all_gather`` of flat shards into
  padded buffers instead of the reference's sequential per-param send/recv loop
  (sharded_ema.py:36-61) — O(1) collectives instead of O(#params) p2p latency.
"""

from __future__ import annotations

from typing import Dict, List, Optional

import torch
import torch.distributed as dist


def partition_by_numel(params: List[torch.Tensor], num_parts: int) -> List[List[int]]:
    """Greedy balanced partition of param indices by numel (largest-first onto
    the lightest part), then each part sorted by original index."""
    order = sorted(range(len(params)), key=lambda i: -params[i].numel())
    loads = [0] * num_parts
    parts: List[List[int]] = [[] for _ in range(num_parts)]
    for i in order:
        j = loads.index(min(loads))
        parts[j].append(i)
        loads[j] += params[i].numel()
    return [sorted(p) for p in parts]


class ShardedEMA:
    def __init__(self, model: torch.nn.Module, decay: float = 0.999,
                 group: Optional[dist.ProcessGroup] = None):
        self.decay = decay
        self.group = group
        self.rank = dist.get_rank(group) if dist.is_initialized() else 0
        self.world = dist.get_world_size(group) if dist.is_initialized() else 1

        self._names: List[str] = []
        self._params: List[torch.Tensor] = []
        for name, p in model.named_parameters():
            if p.requires_grad:
                self._names.append(name)
                self._params.append(p)

        self._parts = partition_by_numel(self._params, self.world)
        self._my_idx = self._parts[self.rank]
        my_numel = sum(self._params[i].numel() for i in self._my_idx)

        dev = self._params[0].device if self._params else torch.device("cpu")
        self._flat = torch.empty(my_numel, dtype=torch.float32, device=dev)
        self._views: Dict[int, torch.Tensor] = {}
        off = 0
        for i in self._my_idx:
            n = self._params[i].numel()
            view = self._flat.narrow(0, off, n)
            view.copy_(self._params[i].detach().float().reshape(-1))
            self._views[i] = view
            off += n

    @torch.no_grad()
    def update(self, decay: Optional[float] = None):
        d = self.decay if decay is None else decay
        if self._flat.numel() == 0:
            return
        # per-param view updates: a single flat cat would materialize a full
        # fp32 copy of the shard (~30 GB on Llama-8B) — OOM territory
        for i in self._my_idx:
            p = self._params[i].detach().reshape(-1)
            v = self._views[i]
            if v.is_cuda and p.dtype in (torch.float32, torch.bfloat16):
                # fused HIP pass (reads bf16 params directly — the lerp_
                # fallback materialized p.float() every step)
                from ..ops import ema_update_
                ema_update_(v, p, d)
            else:
                v.lerp_(p.float(), 1.0 - d)

    @torch.no_grad()
    def state_dict_shard(self) -> Dict[str, torch.Tensor]:
        return {self._names[i]: self._views[i]
                .view(self._params[i].shape).clone()
                for i in self._my_idx}

    @torch.no_grad()
    def state_dict_cpu(self) -> Optional[Dict[str, torch.Tensor]]:
        """Reassemble the full EMA state on group-rank 0 (returns None on other
        ranks).  One flat all_gather of padded shards."""
        if self.world == 1:
            return {self._names[i]: self._views[i].view(self._params[i].shape)
                    .cpu().clone() for i in range(len(self._params))}

        max_numel = max(
            sum(self._params[i].numel() for i in part) for part in self._parts)
        send = torch.zeros(max_numel, dtype=torch.float32,
                           device=self._flat.device)
        send[:self._flat.numel()].copy_(self._flat)
        bufs = [torch.empty_like(send) for _ in range(self.world)]
        dist.all_gather(bufs, send, group=self.group)
        if self.rank != 0:
            return None
        out: Dict[str, torch.Tensor] = {}
        for r, part in enumerate(self._parts):
            off = 0
            for i in part:
                n = self._params[i].numel()
                out[self._names[i]] = bufs[r][off:off + n] \
                    .view(self._params[i].shape).cpu().clone()
                off += n
        return out

    @torch.no_grad()
    def load_state_dict(self, full: Dict[str, torch.Tensor]):
        """Restore from a FULL (dense) EMA state dict, e.g. the
        ``ema_step{N}.pth`` file written by ``save_checkpoint``; each rank
        copies only the params of its own shard."""
        for i in self._my_idx:
            name = self._names[i]
            if name not in full:
                raise KeyError(f"EMA checkpoint is missing param '{name}'")
            self._views[i].copy_(full[name].detach().reshape(-1).to(
                device=self._views[i].device, dtype=torch.float32))

    @torch.no_grad()
    def verify_with_gt(self, gt: Dict[str, torch.Tensor],
                       rtol: float = 1e-6, atol: float = 1e-6) -> bool:
        """Compare this rank's shard against a ground-truth dense EMA dict."""
        for i in self._my_idx:
            name = self._names[i]
            mine = self._views[i].view(self._params[i].shape)
            if not torch.allclose(mine.cpu(), gt[name].float().cpu(),
                                  rtol=rtol, atol=atol):
                return False
        return True
