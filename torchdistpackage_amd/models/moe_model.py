"""Mixtral-style MoE transformer (BASELINE.json config 4: 8-expert MoE with
expert-parallel all-to-all + MoE-DP replicas over xGMI).

GPT-2 backbone with every block's dense MLP replaced by an
ExpertParallelMoE layer; attention/norms identical to models.gpt2.  The
reference's MoE recipe swaps ``block.mlp`` for DeepSpeed's MoE layer
(/root/reference/explore/moe/ds_fmoe_main.py:22-25) — here the dispatch
layer is in-package (moe/layer.py) and the swap is a config flag.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..moe import ExpertParallelMoE
from ..ops import LayerNorm
from ..parallel.tensor import Attention
from .gpt2 import GPT2Config, GPT2Embedding, GPT2Head


@dataclass
class MoEConfig:
    vocab_size: int = 50304
    n_layer: int = 12
    n_head: int = 12
    dim: int = 768
    max_seq: int = 1024
    num_experts: int = 8
    top_k: int = 2
    hidden_mult: int = 4
    aux_loss_weight: float = 0.01
    causal: bool = True
    tie_weights: bool = True


def mixtral_style_8x() -> MoEConfig:
    """8-expert GPT-2-medium-class MoE (per-token params ~ dense medium)."""
    return MoEConfig(n_layer=24, n_head=8, dim=1024, num_experts=8, top_k=2)


class MoEBlock(nn.Module):
    def __init__(self, cfg: MoEConfig, device=None, dtype=None):
        super().__init__()
        kw = {"device": device, "dtype": dtype}
        self.ln_1 = LayerNorm(cfg.dim, **kw)
        self.attn = Attention(cfg.dim, cfg.n_head, causal=cfg.causal, **kw)
        self.ln_2 = LayerNorm(cfg.dim, **kw)
        # batched=False: the per-expert segment path measured FASTER at
        # the moe_8x bench scale (93.7k vs 80.9k tok/s — its narrow()
        # slices are zero-copy; BatchedExperts' gathers cost more than the
        # padding fills they remove, profiles/r02_notes.md)
        self.moe = ExpertParallelMoE(cfg.dim, cfg.num_experts, cfg.top_k,
                                     cfg.hidden_mult, **kw)

    def forward(self, x):
        x = x + self.attn(self.ln_1(x))
        x = x + self.moe(self.ln_2(x))
        return x


class MoEModel(nn.Module):
    def __init__(self, cfg: MoEConfig, device=None, dtype=None):
        super().__init__()
        self.cfg = cfg
        kw = {"device": device, "dtype": dtype}
        g2 = GPT2Config(vocab_size=cfg.vocab_size, n_layer=cfg.n_layer,
                        n_head=cfg.n_head, dim=cfg.dim, max_seq=cfg.max_seq,
                        tie_weights=cfg.tie_weights)
        self.embed = GPT2Embedding(g2, **kw)
        self.blocks = nn.ModuleList(
            [MoEBlock(cfg, **kw) for _ in range(cfg.n_layer)])
        self.head = GPT2Head(g2, self.embed.wte if cfg.tie_weights else None,
                             **kw)

    def forward(self, idx: torch.Tensor,
                labels: Optional[torch.Tensor] = None) -> dict:
        x = self.embed(idx)
        aux_total = None
        for blk in self.blocks:
            x = blk(x)
            aux = blk.moe.aux_loss
            aux_total = aux if aux_total is None else aux_total + aux
        logits = self.head(x)
        out = {"logits": logits, "aux_loss": aux_total}
        if labels is not None:
            from ..ops import cross_entropy_loss
            ce = cross_entropy_loss(logits.transpose(0, 1),
                                    labels.transpose(0, 1))
            out["loss"] = ce + self.cfg.aux_loss_weight * aux_total.to(ce.dtype)
        return out

    def expert_parameters(self):
        for p in self.parameters():
            if getattr(p, "expert_parallel", False):
                yield p

    def non_expert_parameters(self):
        for p in self.parameters():
            if not getattr(p, "expert_parallel", False):
                yield p
