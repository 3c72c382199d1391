"""MLP and tensor-parallel MLP blocks.

Reference parity: /root/reference/torchdistpackage/parallel/tensor_parallel/
mlp.py (Mlp/TpMlp: fc1 Col -> GELU -> fc2 Row, SP gather on entry when the
input is sequence-parallel).

Layout convention (Megatron/MI355X): activations are sequence-first
(S, B, D) inside TP blocks so SP shards dim 0 and the RCCL
all-gather/reduce-scatter run on contiguous memory.  GELU is fused into the
fc1 epilogue by the in-tree HIP bias_gelu kernel on GPU (ops.bias_gelu),
saving one full activation round-trip to HBM3E.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from .tp_utils import (ColParallelLinear, RowParallelLinear, TpLinear,
                       get_tp_size, is_sequence_parallel,
                       maybe_gather_for_sequence_parallel)
from ...ops import bias_gelu


class Mlp(nn.Module):
    """Non-parallel reference MLP (oracle for TP tests)."""

    def __init__(self, dim: int, hidden_mult: int = 4, bias: bool = True,
                 device=None, dtype=None):
        super().__init__()
        self.fc1 = TpLinear(dim, dim * hidden_mult, bias=bias,
                            device=device, dtype=dtype)
        self.fc2 = TpLinear(dim * hidden_mult, dim, bias=bias,
                            device=device, dtype=dtype)

    def forward(self, x):
        return self.fc2(F.gelu(self.fc1(x), approximate="tanh"))


class TpMlp(nn.Module):
    """Tensor-parallel MLP: Col(fc1) -> fused bias+GELU -> Row(fc2).

    If the input is SP-tagged, gathers the full sequence on entry; fc2
    reduce-scatters back into SP when ``sequence_parallel``.
    """

    def __init__(self, dim: int, hidden_mult: int = 4, bias: bool = True,
                 dropout: float = 0.0, sequence_parallel: bool = False,
                 device=None, dtype=None):
        super().__init__()
        self.dropout = dropout
        self.fc1 = ColParallelLinear(dim, dim * hidden_mult, bias=bias,
                                     device=device, dtype=dtype)
        self.fc2 = RowParallelLinear(dim * hidden_mult, dim, bias=bias,
                                     sequence_parallel=sequence_parallel,
                                     device=device, dtype=dtype)

    def forward(self, x):
        from .tp_utils import copy_to_tp_region, \
            gather_from_sequence_parallel_region
        if is_sequence_parallel(x):
            # SP entry: gather fwd / reduce-scatter bwd covers the input-grad
            # reduction that copy_to_tp would otherwise do
            x = gather_from_sequence_parallel_region(x)
        else:
            x = copy_to_tp_region(x)
        from ...ops.gemm import linear as fast_linear
        h = fast_linear(x, self.fc1.weight)  # bias deferred to fused kernel
        h = bias_gelu(h, self.fc1.bias)
        if self.dropout > 0 and self.training:
            h = F.dropout(h, p=self.dropout)
        return self.fc2(h)

    @torch.no_grad()
    def init_weight_from_full(self, full_mlp: Mlp):
        self.fc1.init_weight_from_full(full_mlp.fc1.weight,
                                       full_mlp.fc1.bias)
        self.fc2.init_weight_from_full(full_mlp.fc2.weight,
                                       full_mlp.fc2.bias)
