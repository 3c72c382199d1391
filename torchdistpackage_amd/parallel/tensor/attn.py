"""Attention and tensor-parallel attention.

Reference parity: /root/reference/torchdistpackage/parallel/tensor_parallel/
attn.py (Attention/TpAttention: fused-QKV 3*dim ColParallel with
heads-per-partition = nh/tp, RowParallel out-proj).  Where the reference runs
naive math attention (attn.py:85-88: q@k^T -> softmax -> @v), this stack runs
the in-tree gfx950 flash-attention kernel (ops.flash_attention; blockwise
online-softmax per the reference spec explore/flash-attn/tile_attn.py).

Layout: activations are sequence-first (S, B, D); SP shards dim 0.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ...ops import fused_qkv_attention
from .tp_utils import (ColParallelLinear, RowParallelLinear, TpLinear,
                       copy_to_tp_region, gather_from_sequence_parallel_region,
                       get_tp_size, is_sequence_parallel)


def _cached_sdpa(q: torch.Tensor, k_cache: torch.Tensor,
                 v_cache: torch.Tensor, pos0: int,
                 causal: bool) -> torch.Tensor:
    """Attention of q (B, H, S_new, hd) against cache positions
    [0, pos0+S_new), with the causal mask aligned so new position i sees
    cached positions <= pos0 + i (torch's ``is_causal`` is top-left
    aligned, wrong for a chunk appended mid-sequence)."""
    S = q.shape[2]
    ka = k_cache[:, :, :pos0 + S]
    va = v_cache[:, :, :pos0 + S]
    if not causal or S == 1:
        return F.scaled_dot_product_attention(q, ka, va)
    if pos0 == 0:
        return F.scaled_dot_product_attention(q, ka, va, is_causal=True)
    mask = torch.arange(pos0 + S, device=q.device)[None, :] <= \
        (pos0 + torch.arange(S, device=q.device))[:, None]
    return F.scaled_dot_product_attention(q, ka, va, attn_mask=mask)


def _sdpa(x_qkv: torch.Tensor, n_head: int, causal: bool) -> torch.Tensor:
    """(S, B, 3*Hl*hd) fused qkv -> (S, B, Hl*hd) attention output.

    Zero-copy path: the flash kernels read/write strided views into the fused
    qkv / output buffers (no permute-contiguous transposes).
    """
    return fused_qkv_attention(x_qkv.contiguous(), n_head, causal=causal)


class Attention(nn.Module):
    """Non-parallel attention (oracle for TP tests)."""

    def __init__(self, dim: int, n_head: int, bias: bool = True,
                 causal: bool = True, device=None, dtype=None):
        super().__init__()
        assert dim % n_head == 0
        self.n_head = n_head
        self.causal = causal
        self.qkv = TpLinear(dim, 3 * dim, bias=bias, device=device, dtype=dtype)
        self.proj = TpLinear(dim, dim, bias=bias, device=device, dtype=dtype)

    def forward(self, x):
        return self.proj(_sdpa(self.qkv(x), self.n_head, self.causal))


class TpAttention(nn.Module):
    """Tensor-parallel attention: heads split over TP ranks.

    qkv is ColParallel over 3*dim (each rank computes its nh/tp heads'
    Q,K,V); out-proj is RowParallel ending in all-reduce (or reduce-scatter
    into SP).  SP-tagged inputs are gathered on entry.
    """

    def __init__(self, dim: int, n_head: int, bias: bool = True,
                 causal: bool = True, sequence_parallel: bool = False,
                 device=None, dtype=None):
        super().__init__()
        tp = get_tp_size()
        assert dim % n_head == 0 and n_head % tp == 0, (dim, n_head, tp)
        self.n_head = n_head
        self.n_head_local = n_head // tp
        self.causal = causal
        self.qkv = ColParallelLinear(dim, 3 * dim, bias=bias,
                                     device=device, dtype=dtype)
        self.proj = RowParallelLinear(dim, dim, bias=bias,
                                      sequence_parallel=sequence_parallel,
                                      device=device, dtype=dtype)

    def forward(self, x):
        if is_sequence_parallel(x):
            x = gather_from_sequence_parallel_region(x)
        else:
            x = copy_to_tp_region(x)
        from ...ops.gemm import linear as fast_linear
        qkv = fast_linear(x, self.qkv.weight, self.qkv.bias)
        # qkv layout per rank: [q_local | k_local | v_local] thanks to the
        # interleaved loader (init_qkv_weight_from_full) / native init
        o = _sdpa(qkv, self.n_head_local, self.causal)
        return self.proj(o)

    @torch.no_grad()
    def decode_step(self, x: torch.Tensor, k_cache: torch.Tensor,
                    v_cache: torch.Tensor, pos0: int) -> torch.Tensor:
        """KV-cache inference step (tp=1 only; see inference/generate.py).

        x (S_new, B, D) — the new tokens' hidden states; writes their K/V
        into ``k_cache``/``v_cache`` (B, H, max_seq, hd) at ``pos0`` and
        attends q against positions [0, pos0+S_new).  Decode is
        memory-bound GEMV-shaped work, so eager SDPA over the cache is the
        right tool (the flash kernel is the training-shape path).
        """
        from ...ops.gemm import linear as fast_linear
        S, B, D = x.shape
        hl = self.n_head_local
        hd = D // self.n_head
        qkv = fast_linear(x, self.qkv.weight, self.qkv.bias)
        q, k, v = qkv.split(hl * hd, dim=-1)

        def v4(t):
            return t.reshape(S, B, hl, hd).permute(1, 2, 0, 3)

        q, k, v = v4(q), v4(k), v4(v)
        k_cache[:, :, pos0:pos0 + S] = k
        v_cache[:, :, pos0:pos0 + S] = v
        o = _cached_sdpa(q, k_cache, v_cache, pos0, self.causal)
        o = o.permute(2, 0, 1, 3).reshape(S, B, hl * hd)
        return self.proj(o)

    @torch.no_grad()
    def init_from_full(self, full_attn: Attention):
        self.qkv.init_qkv_weight_from_full(full_attn.qkv.weight,
                                           full_attn.qkv.bias, num_splits=3)
        self.proj.init_weight_from_full(full_attn.proj.weight,
                                        full_attn.proj.bias)
