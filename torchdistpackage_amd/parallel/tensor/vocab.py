"""Vocab-parallel embedding, LM head and cross-entropy (Megatron-style).

Not in the reference (its TP blocks stop at the transformer body;
examples compute the full-vocab head + CE replicated on every TP rank,
which VERDICT r01 flags as the single biggest replicated cost of the
tp2 flagship config).  Here the vocab dimension is sharded over the TP
group:

- ``VocabParallelEmbedding``: each rank holds ``V/tp`` rows; out-of-shard
  token ids are masked to 0 and the partial lookups are summed with ONE
  all-reduce (or reduce-scattered straight into the SP layout when
  ``sequence_parallel`` — one collective instead of all-reduce + split).
- ``VocabParallelHead``: column-parallel projection producing LOCAL logits
  ``(.., V/tp)`` — never gathered.
- ``vocab_parallel_cross_entropy``: CE over vocab-sharded logits.  Instead
  of gathering 50k-wide logits, each rank computes its shard's online LSE
  and the local target logit; ONE all-gather of per-token LSEs (tp x N
  floats) + ONE all-reduce of target logits (N floats) completes the loss.
  Backward needs no communication: d(logits_local) = softmax_local - onehot.

On xGMI, the collectives this trades away (full-vocab logits were
(B*S, V) bf16 per rank per step) dwarf the two (N,)-float exchanges added.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F

from .tp_utils import (get_tp_group, get_tp_rank, get_tp_size,
                       reduce_scatter_to_sequence_parallel_region,
                       reduce_from_tp_region)


def _vocab_range(vocab_size: int, rank: int, world: int):
    assert vocab_size % world == 0, (vocab_size, world)
    per = vocab_size // world
    return rank * per, (rank + 1) * per


class VocabParallelEmbedding(nn.Module):
    """Embedding with the vocab (num_embeddings) dim sharded over TP.

    Weights are initialized by drawing the FULL (V, D) table with the
    current RNG and slicing this rank's rows, so TP ranks hold consistent
    shards of one well-defined full table (and tests can compare against a
    tp=1 oracle seeded identically).
    """

    def __init__(self, num_embeddings: int, embedding_dim: int,
                 init_std: float = 0.02, device=None, dtype=None):
        super().__init__()
        self.num_embeddings = num_embeddings
        self.embedding_dim = embedding_dim
        tp, rank = get_tp_size(), get_tp_rank()
        self.vocab_start, self.vocab_end = _vocab_range(
            num_embeddings, rank, tp)
        full = torch.empty(num_embeddings, embedding_dim, device=device,
                           dtype=dtype)
        nn.init.normal_(full, std=init_std)
        self.weight = nn.Parameter(
            full[self.vocab_start:self.vocab_end].clone())
        del full

    def forward(self, idx: torch.Tensor,
                sequence_parallel_out: bool = False) -> torch.Tensor:
        """idx (B, S) -> (B, S, D) summed over shards; if
        ``sequence_parallel_out``, returns the (S/tp, B, D) SP shard
        (callers transpose to seq-first BEFORE the reduce-scatter)."""
        tp = get_tp_size()
        if tp == 1:
            out = F.embedding(idx, self.weight)
            return out
        mask = (idx < self.vocab_start) | (idx >= self.vocab_end)
        local_idx = (idx - self.vocab_start).masked_fill(mask, 0)
        out = F.embedding(local_idx, self.weight)
        out = out.masked_fill(mask.unsqueeze(-1), 0.0)
        if sequence_parallel_out:
            # (B, S, D) -> (S, B, D) then reduce-scatter along seq: one
            # collective produces the SP shard directly
            out = out.transpose(0, 1).contiguous()
            return reduce_scatter_to_sequence_parallel_region(out)
        return reduce_from_tp_region(out)

    @torch.no_grad()
    def load_from_full(self, full_weight: torch.Tensor):
        self.weight.copy_(full_weight[self.vocab_start:self.vocab_end])


class VocabParallelHead(nn.Module):
    """LM head projecting (.., D) -> LOCAL logits (.., V/tp); pair with
    ``vocab_parallel_cross_entropy``.  Pass ``weight`` to tie with a
    ``VocabParallelEmbedding`` shard."""

    def __init__(self, dim: int, vocab_size: int,
                 weight: Optional[nn.Parameter] = None,
                 init_std: float = 0.02, device=None, dtype=None):
        super().__init__()
        self.vocab_size = vocab_size
        tp, rank = get_tp_size(), get_tp_rank()
        self.vocab_start, self.vocab_end = _vocab_range(vocab_size, rank, tp)
        if weight is not None:
            self.weight = weight
        else:
            full = torch.empty(vocab_size, dim, device=device, dtype=dtype)
            nn.init.normal_(full, std=init_std)
            self.weight = nn.Parameter(
                full[self.vocab_start:self.vocab_end].clone())
            del full

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return F.linear(x, self.weight)


class _VocabParallelCE(torch.autograd.Function):
    """Mean CE over (N, V/tp) vocab-sharded logits.

    fp32 math throughout (matches the fused non-TP kernel's accumulate).
    """

    @staticmethod
    def forward(ctx, logits: torch.Tensor, target: torch.Tensor,
                vocab_start: int, vocab_end: int):
        group = get_tp_group()
        tp = get_tp_size()
        N = logits.shape[0]
        fused = logits.is_cuda and logits.dtype == torch.bfloat16
        if fused:
            # ONE online pass over the (N, V/tp) shard (ops ce_partial_fwd):
            # no fp32 logits materialization, no separate max/exp passes
            from ...ops import ext
            local_t = torch.where(
                (target >= vocab_start) & (target < vocab_end),
                target - vocab_start, torch.full_like(target, -1))
            lse_local, tgt_logit = ext("ce_partial").ce_partial_fwd(
                logits, local_t)
        else:
            x = logits.float()
            local_max = x.max(dim=-1).values                  # (N,)
            lse_local = torch.log(
                torch.exp(x - local_max.unsqueeze(-1)).sum(-1)) + local_max
            in_range = (target >= vocab_start) & (target < vocab_end)
            local_t = (target - vocab_start).masked_fill(~in_range, 0)
            tgt_logit = x.gather(-1, local_t.unsqueeze(-1)).squeeze(-1)
            tgt_logit = tgt_logit.masked_fill(~in_range, 0.0)
        if tp > 1:
            all_lse = torch.empty(tp * N, dtype=lse_local.dtype,
                                  device=lse_local.device)
            dist.all_gather_into_tensor(all_lse, lse_local.contiguous(),
                                        group=group)
            lse = torch.logsumexp(all_lse.reshape(tp, N), dim=0)
            dist.all_reduce(tgt_logit, op=dist.ReduceOp.SUM, group=group)
        else:
            lse = lse_local
        loss = (lse - tgt_logit).mean()
        ctx.save_for_backward(logits, target, lse)
        ctx.vocab_start, ctx.vocab_end = vocab_start, vocab_end
        return loss

    @staticmethod
    def backward(ctx, gout):
        logits, target, lse = ctx.saved_tensors
        N = logits.shape[0]
        if logits.is_cuda and logits.dtype == torch.bfloat16:
            # ce_bwd with the GLOBAL lse: out-of-shard targets (-1) never
            # match the one-hot test, so the kernel is reused verbatim
            from ...ops import ext
            local_t = torch.where(
                (target >= ctx.vocab_start) & (target < ctx.vocab_end),
                target - ctx.vocab_start, torch.full_like(target, -1))
            grad = ext("ce_partial").ce_bwd(
                logits, local_t, lse.contiguous(),
                gout.reshape(1).float().contiguous())
            return grad, None, None, None
        grad = torch.exp(logits.float() - lse.unsqueeze(-1))
        in_range = (target >= ctx.vocab_start) & (target < ctx.vocab_end)
        local_t = (target - ctx.vocab_start).masked_fill(~in_range, 0)
        one = torch.zeros_like(grad)
        one.scatter_(-1, local_t.unsqueeze(-1),
                     in_range.unsqueeze(-1).to(grad.dtype))
        grad = (grad - one) * (gout.float() / N)
        return grad.to(logits.dtype), None, None, None


def vocab_parallel_cross_entropy(logits_local: torch.Tensor,
                                 target: torch.Tensor,
                                 vocab_start: int,
                                 vocab_end: int) -> torch.Tensor:
    """Mean CE over vocab-sharded logits (.., V/tp); flattens leading dims."""
    l2 = logits_local.reshape(-1, logits_local.size(-1))
    t = target.reshape(-1)
    return _VocabParallelCE.apply(l2.contiguous(), t, vocab_start, vocab_end)
