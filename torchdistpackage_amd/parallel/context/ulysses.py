"""Ulysses-style context parallelism: all-to-all head scatter.

Long-sequence scaling the reference lacks entirely ("CP /
ring attention / Ulysses: absent").  Each CP rank holds a sequence shard
(S/cp); one all-to-all converts to a head shard (H/cp heads, FULL sequence),
local flash attention runs over the full sequence, and the inverse all-to-all
restores the sequence shard.  Two all-to-alls per call — on one 8×MI355X node
each rides direct xGMI links.

Requires H % cp == 0 and S % cp == 0; causal masking works unchanged (each
head sees the full sequence).
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist

from ...ops import flash_attention


def _a2a_even(x: torch.Tensor, group) -> torch.Tensor:
    """all_to_all over dim 0 (world-sized leading dim); gloo fallback."""
    world = dist.get_world_size(group)
    x = x.contiguous()
    if dist.get_backend(group) == "gloo":
        # gloo has no alltoall: emulate with all_gather + select (CPU tests)
        me = dist.get_rank(group)
        gathered = [torch.empty_like(x) for _ in range(world)]
        dist.all_gather(gathered, x, group=group)
        chunk = x.shape[0] // world
        return torch.cat([g[me * chunk:(me + 1) * chunk] for g in gathered],
                         dim=0)
    out = torch.empty_like(x)
    dist.all_to_all_single(out, x, group=group)
    return out


class _A2A(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return _a2a_even(x, group)

    @staticmethod
    def backward(ctx, g):
        # all_to_all is its own inverse for even splits
        return _a2a_even(g.contiguous(), ctx.group), None


def _scatter_heads(t: torch.Tensor, cp: int, group) -> torch.Tensor:
    """(B, H, S_loc, D) seq-shard -> (B, H/cp, S, D) head-shard."""
    B, H, S_loc, D = t.shape
    # send peer p: heads [p*H/cp, (p+1)*H/cp) of my S_loc rows
    x = t.reshape(B, cp, H // cp, S_loc, D).permute(1, 0, 2, 3, 4).contiguous()
    x = _A2A.apply(x, group)              # (cp, B, H/cp, S_loc, D) received
    # received chunk p holds peer p's S_loc rows of MY head group
    return x.permute(1, 2, 0, 3, 4).reshape(B, H // cp, cp * S_loc, D)


def _gather_heads(t: torch.Tensor, cp: int, group) -> torch.Tensor:
    """(B, H/cp, S, D) head-shard -> (B, H, S_loc, D) seq-shard."""
    B, Hl, S, D = t.shape
    S_loc = S // cp
    x = t.reshape(B, Hl, cp, S_loc, D).permute(2, 0, 1, 3, 4).contiguous()
    x = _A2A.apply(x, group)              # (cp, B, H/cp, S_loc, D)
    return x.permute(1, 0, 2, 3, 4).reshape(B, cp * Hl, S_loc, D)


def ulysses_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                      causal: bool = True, scale: Optional[float] = None,
                      group: Optional[dist.ProcessGroup] = None):
    """q/k/v: (B, H, S_local, D) sequence shards (rank r holds rows
    [r*S_loc, (r+1)*S_loc)); returns the (B, H, S_local, D) output shard."""
    if group is None or not dist.is_initialized() \
            or dist.get_world_size(group) == 1:
        return flash_attention(q, k, v, causal=causal, scale=scale)
    cp = dist.get_world_size(group)
    H = q.shape[1]
    assert H % cp == 0, (H, cp)
    qh = _scatter_heads(q, cp, group)
    kh = _scatter_heads(k, cp, group)
    vh = _scatter_heads(v, cp, group)
    oh = flash_attention(qh, kh, vh, causal=causal, scale=scale)
    return _gather_heads(oh, cp, group)


class UlyssesAttention(torch.nn.Module):
    def __init__(self, causal: bool = True,
                 group: Optional[dist.ProcessGroup] = None):
        super().__init__()
        self.causal = causal
        self.group = group

    def forward(self, q, k, v):
        return ulysses_attention(q, k, v, causal=self.causal,
                                 group=self.group)
