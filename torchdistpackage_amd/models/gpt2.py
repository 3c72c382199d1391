"""GPT-2 model family on the TP/SP blocks (works at tp=1 as plain blocks).

This is the flagship training model for the benchmarks (BASELINE.json:
GPT-2 small pure-DP; GPT-2 1.3B DP2xPP2xTP2).  The reference has no model
zoo — its examples build throwaway transformers
(/root/reference/examples/model_parallel/test_transformer.py:13-45); here the
zoo is a first-class component.

Layout: embeddings produce (B, S, D); blocks run sequence-first (S, B, D);
loss is fp32 cross-entropy.  Hot ops (LayerNorm, flash attention, bias+GELU)
are the in-tree gfx950 HIP kernels; GEMMs ride hipBLASLt via F.linear.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import LayerNorm
from ..ops.gemm import linear as fast_linear
from ..parallel.tensor import ParallelBlock, get_tp_size


@dataclass
class GPT2Config:
    vocab_size: int = 50304      # padded 50257 for GEMM-friendly vocab
    n_layer: int = 12
    n_head: int = 12
    dim: int = 768
    max_seq: int = 1024
    hidden_mult: int = 4
    dropout: float = 0.0
    causal: bool = True
    sequence_parallel: bool = True
    tie_weights: bool = True
    checkpoint_activations: bool = False  # recompute blocks in backward
    # shard the vocab dim of embedding + head over TP and use vocab-parallel
    # CE (Megatron-style) — at tp>1 this stops every TP rank computing the
    # full 50k-vocab head GEMM + CE (VERDICT r01 missing #5); inert at tp=1
    vocab_parallel: bool = True


def gpt2_small() -> GPT2Config:
    return GPT2Config(n_layer=12, n_head=12, dim=768)


def gpt2_medium() -> GPT2Config:
    return GPT2Config(n_layer=24, n_head=16, dim=1024)


def gpt2_xl_1p3b() -> GPT2Config:
    """GPT-2/3 1.3B-class: 24 layers, d=2048, 16 heads (head_dim 128)."""
    return GPT2Config(n_layer=24, n_head=16, dim=2048)


class GPT2Embedding(nn.Module):
    def __init__(self, cfg: GPT2Config, device=None, dtype=None):
        super().__init__()
        kw = {"device": device, "dtype": dtype}
        self.vocab_parallel = cfg.vocab_parallel and get_tp_size() > 1
        self.sequence_parallel = cfg.sequence_parallel
        if self.vocab_parallel:
            from ..parallel.tensor.vocab import VocabParallelEmbedding
            self.wte = VocabParallelEmbedding(cfg.vocab_size, cfg.dim,
                                              init_std=0.02, **kw)
        else:
            self.wte = nn.Embedding(cfg.vocab_size, cfg.dim, **kw)
            nn.init.normal_(self.wte.weight, std=0.02)
        self.wpe = nn.Embedding(cfg.max_seq, cfg.dim, **kw)
        nn.init.normal_(self.wpe.weight, std=0.01)
        if self.vocab_parallel and self.sequence_parallel:
            # wpe is applied to the LOCAL sequence shard only: its grads
            # need the SP all-reduce over TP
            from ..parallel.tensor import mark_sequence_parallel_params
            mark_sequence_parallel_params(self.wpe)

    def forward(self, idx: torch.Tensor) -> torch.Tensor:
        # idx (B, S) -> hidden (S, B, D) sequence-first for the TP blocks
        # (SP shard (S/tp, B, D), tagged, on the vocab-parallel + SP path)
        B, S = idx.shape
        pos = torch.arange(S, device=idx.device)
        if self.vocab_parallel:
            from ..parallel.tensor import set_sequence_parallel_attr
            if self.sequence_parallel and get_tp_size() > 1:
                from ..parallel.tensor import get_tp_rank
                # partial lookup reduce-scattered straight into the SP
                # layout (one collective); wpe added on the local shard
                x = self.wte(idx, sequence_parallel_out=True)  # (S/tp, B, D)
                shard = S // get_tp_size()
                r = get_tp_rank()
                x = x + self.wpe(pos[r * shard:(r + 1) * shard])[:, None, :]
                set_sequence_parallel_attr(x)
                return x
            x = self.wte(idx) + self.wpe(pos)[None, :, :]
            return x.transpose(0, 1).contiguous()
        x = self.wte(idx) + self.wpe(pos)[None, :, :]
        return x.transpose(0, 1).contiguous()


class GPT2Head(nn.Module):
    def __init__(self, cfg: GPT2Config, wte,
                 device=None, dtype=None):
        super().__init__()
        kw = {"device": device, "dtype": dtype}
        self.ln_f = LayerNorm(cfg.dim, **kw)
        self.vocab_size = cfg.vocab_size
        self.vocab_parallel = cfg.vocab_parallel and get_tp_size() > 1
        self.sequence_parallel = cfg.sequence_parallel
        if self.vocab_parallel:
            from ..parallel.tensor.vocab import VocabParallelHead
            tied = wte.weight if (cfg.tie_weights and wte is not None) \
                else None
            vh = VocabParallelHead(cfg.dim, cfg.vocab_size, weight=tied,
                                   init_std=0.02, **kw)
            self.vocab_start, self.vocab_end = vh.vocab_start, vh.vocab_end
            self.weight = tied if tied is not None else vh.weight
            if self.sequence_parallel:
                # ln_f runs on the SP sequence shard in this mode
                from ..parallel.tensor import mark_sequence_parallel_params
                mark_sequence_parallel_params(self.ln_f)
        elif cfg.tie_weights and wte is not None:
            self.weight = wte.weight  # shared Parameter
        else:
            self.weight = nn.Parameter(
                torch.empty(cfg.vocab_size, cfg.dim, **kw))
            nn.init.normal_(self.weight, std=0.02)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        # x (S, B, D) [or SP shard (S/tp, B, D)] -> logits (B, S, V)
        # (LOCAL (B, S, V/tp) logits when vocab_parallel — pair with
        # vocab_parallel_cross_entropy, never gathered)
        from ..parallel.tensor import (is_sequence_parallel,
                                       gather_from_sequence_parallel_region,
                                       copy_to_tp_region)
        if self.vocab_parallel:
            sp_in = is_sequence_parallel(x)  # ln_f output loses the tag
            x = self.ln_f(x)
            if sp_in and get_tp_size() > 1:
                # bwd reduce-scatter: each rank's dx covers only its vocab
                # shard; the sum over TP happens on the way back to SP
                x = gather_from_sequence_parallel_region(
                    x, bwd_mode="reduce_scatter")
            else:
                # replicated activations: dx partial per rank -> bwd
                # all-reduce via the copy region
                x = copy_to_tp_region(x)
            logits = fast_linear(x, self.weight)
            return logits.transpose(0, 1)
        x = self.ln_f(x)
        logits = fast_linear(x, self.weight)
        return logits.transpose(0, 1)


class GPT2Model(nn.Module):
    def __init__(self, cfg: GPT2Config, device=None, dtype=None):
        super().__init__()
        self.cfg = cfg
        kw = {"device": device, "dtype": dtype}
        self.embed = GPT2Embedding(cfg, **kw)
        self.blocks = nn.ModuleList([
            ParallelBlock(cfg.dim, cfg.n_head, cfg.hidden_mult,
                          causal=cfg.causal, dropout=cfg.dropout,
                          sequence_parallel=cfg.sequence_parallel, **kw)
            for _ in range(cfg.n_layer)])
        self.head = GPT2Head(cfg, self.embed.wte if cfg.tie_weights else None,
                             **kw)

    def forward(self, idx: torch.Tensor,
                labels: Optional[torch.Tensor] = None) -> dict:
        x = self.embed(idx)
        use_ckpt = self.cfg.checkpoint_activations and self.training \
            and torch.is_grad_enabled()
        for blk in self.blocks:
            if use_ckpt:
                from torch.utils.checkpoint import checkpoint
                x = checkpoint(blk, x, use_reentrant=False)
            else:
                x = blk(x)
        from ..parallel.tensor import (is_sequence_parallel,
                                       gather_from_sequence_parallel_region)
        if self.head.vocab_parallel:
            # head gathers SP internally (bwd reduce-scatter) and emits
            # LOCAL (B, S, V/tp) logits — the full-vocab head GEMM + CE is
            # never replicated across TP ranks
            logits = self.head(x)
            out = {"logits": logits}
            if labels is not None:
                from ..parallel.tensor.vocab import \
                    vocab_parallel_cross_entropy
                out["loss"] = vocab_parallel_cross_entropy(
                    logits.transpose(0, 1), labels.transpose(0, 1),
                    self.head.vocab_start, self.head.vocab_end)
            return out
        if is_sequence_parallel(x) and get_tp_size() > 1:
            # split-backward: the head + loss are computed identically on
            # every TP rank, so each rank's grad is already complete
            x = gather_from_sequence_parallel_region(x, bwd_mode="split")
        logits = self.head(x)
        out = {"logits": logits}
        if labels is not None:
            from ..ops import cross_entropy_loss
            # logits is a (B,S,V) transpose VIEW of the contiguous (S,B,V)
            # head output; transposing back avoids a reshape copy in the loss
            out["loss"] = cross_entropy_loss(logits.transpose(0, 1),
                                             labels.transpose(0, 1))
        return out

    # -- pipeline partitioning support ----------------------------------

    def to_stage_layers(self) -> List[nn.Module]:
        """Flat layer list for the pipeline partitioner: [embed, blocks...,
        head].  Stages feed (S, B, D) activations between each other."""
        return [self.embed, *self.blocks, self.head]

    @torch.no_grad()
    def num_params(self, non_embedding: bool = False) -> int:
        n = sum(p.numel() for p in self.parameters())
        if non_embedding:
            n -= self.embed.wpe.weight.numel()
        return n
