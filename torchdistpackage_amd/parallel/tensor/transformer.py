"""Transformer blocks: plain and tensor+sequence-parallel.

Reference parity: /root/reference/torchdistpackage/parallel/tensor_parallel/
transformer.py (Block/ParallelBlock pre-LN residual x2; LayerNorm computed on
the SP-split sequence shard — the Megatron-SP memory saving; ParallelBlock
.init_from_full weight surgery; Transformer = depth x blocks + final SP
gather).

LayerNorm runs on the in-tree HIP kernel (ops.LayerNorm); activations are
sequence-first (S, B, D).
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ...ops import LayerNorm
from .attn import Attention, TpAttention
from .mlp import Mlp, TpMlp
from .tp_utils import (gather_from_sequence_parallel_region, get_tp_size,
                       is_sequence_parallel,
                       maybe_split_into_sequence_parallel,
                       set_sequence_parallel_attr)


class Block(nn.Module):
    """Pre-LN transformer block (oracle for TP/SP tests)."""

    def __init__(self, dim: int, n_head: int, hidden_mult: int = 4,
                 bias: bool = True, causal: bool = True,
                 device=None, dtype=None):
        super().__init__()
        kw = {"device": device, "dtype": dtype}
        self.ln_1 = LayerNorm(dim, **kw)
        self.attn = Attention(dim, n_head, bias=bias, causal=causal, **kw)
        self.ln_2 = LayerNorm(dim, **kw)
        self.mlp = Mlp(dim, hidden_mult, bias=bias, **kw)

    def forward(self, x):
        x = x + self.attn(self.ln_1(x))
        x = x + self.mlp(self.ln_2(x))
        return x


class ParallelBlock(nn.Module):
    """TP(+SP) transformer block.

    With ``sequence_parallel``: x enters/leaves as an SP shard (S/tp, B, D);
    LN + residual run on the shard (memory / tp_size); attn and mlp gather
    internally and reduce-scatter back.
    Without SP: x is replicated; attn/mlp end in all-reduce.
    """

    def __init__(self, dim: int, n_head: int, hidden_mult: int = 4,
                 bias: bool = True, causal: bool = True, dropout: float = 0.0,
                 sequence_parallel: bool = True, device=None, dtype=None):
        super().__init__()
        kw = {"device": device, "dtype": dtype}
        self.sequence_parallel = sequence_parallel and get_tp_size() > 1
        self.dropout = dropout
        self.ln_1 = LayerNorm(dim, **kw)
        self.attn = TpAttention(dim, n_head, bias=bias, causal=causal,
                                sequence_parallel=self.sequence_parallel, **kw)
        self.ln_2 = LayerNorm(dim, **kw)
        self.mlp = TpMlp(dim, hidden_mult, bias=bias, dropout=dropout,
                         sequence_parallel=self.sequence_parallel, **kw)
        if self.sequence_parallel:
            # LN grads come from the local sequence shard -> all-reduce over
            # TP needed once per iteration (allreduce_sequence_parallel_grads)
            from .tp_utils import mark_sequence_parallel_params
            mark_sequence_parallel_params(self.ln_1)
            mark_sequence_parallel_params(self.ln_2)

    def forward(self, x):
        if self.sequence_parallel:
            x = maybe_split_into_sequence_parallel(x)
        h = self.ln_1(x)
        if self.sequence_parallel:
            set_sequence_parallel_attr(h)
        a = self.attn(h)
        if self.dropout > 0 and self.training:
            import torch.nn.functional as F
            a = F.dropout(a, p=self.dropout)
        x = x + a
        h = self.ln_2(x)
        if self.sequence_parallel:
            set_sequence_parallel_attr(h)
        x = x + self.mlp(h)
        if self.sequence_parallel:
            set_sequence_parallel_attr(x)
        return x

    @torch.no_grad()
    def decode_step(self, x, k_cache, v_cache, pos0: int):
        """KV-cache inference step (tp=1 only; see inference/generate.py):
        the training forward's SP plumbing is inert at tp=1, so LN/MLP are
        reused as-is and only attention takes the cache."""
        x = x + self.attn.decode_step(self.ln_1(x), k_cache, v_cache, pos0)
        return x + self.mlp(self.ln_2(x))

    @torch.no_grad()
    def init_from_full(self, full: Block):
        self.ln_1.load_state_dict(full.ln_1.state_dict())
        self.ln_2.load_state_dict(full.ln_2.state_dict())
        self.attn.init_from_full(full.attn)
        self.mlp.init_weight_from_full(full.mlp)


class Transformer(nn.Module):
    """depth x blocks (+ final SP gather), reference transformer.py:88-100."""

    def __init__(self, dim: int, n_head: int, depth: int,
                 hidden_mult: int = 4, parallel: bool = True,
                 causal: bool = True, sequence_parallel: bool = True,
                 device=None, dtype=None):
        super().__init__()
        cls = ParallelBlock if parallel else Block
        kw = dict(bias=True, causal=causal, device=device, dtype=dtype)
        if parallel:
            kw["sequence_parallel"] = sequence_parallel
        self.blocks = nn.ModuleList(
            [cls(dim, n_head, hidden_mult, **kw) for _ in range(depth)])

    def forward(self, x):
        for blk in self.blocks:
            x = blk(x)
        if is_sequence_parallel(x) and get_tp_size() > 1:
            # final gather feeds replicated compute (loss/head computed
            # identically on every TP rank): each rank's backward grad is
            # already the full gradient, so backward takes the local slice
            # (reference tp_utils.py:126-149 'already summed' mode) — NOT
            # reduce-scatter, which would double-count
            x = gather_from_sequence_parallel_region(x, bwd_mode="split")
        return x
