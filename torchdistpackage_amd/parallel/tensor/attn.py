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
    def init_from_full(self, full_attn: Attention):
        self.qkv.init_qkv_weight_from_full(full_attn.qkv.weight,
                                           full_attn.qkv.bias, num_splits=3)
        self.proj.init_weight_from_full(full_attn.proj.weight,
                                        full_attn.proj.bias)
