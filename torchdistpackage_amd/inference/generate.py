"""KV-cache autoregressive generation for the model zoo (single-GPU serving).

The reference has no inference path at all (its ``forward_eval`` pipelines
full forwards, pipeline_sched.py:233-269); this module adds the serving-side
basics: prefill + incremental decode with per-layer K/V caches, greedy and
temperature/top-k sampling.

Scope: tp=1 (single device).  At tp=1 the Col/Row parallel linears, SP
plumbing and vocab-parallel head are all inert, so the training modules'
weights are reused directly by the ``decode_step`` methods
(parallel/tensor/attn.py, models/llama.py).  Decode-shaped attention is
GEMV-like and memory-bound, so it runs eager SDPA over the cache rather
than the training flash kernel.
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import torch

from ..parallel.tensor import get_tp_size


def _alloc_caches(n_layer: int, batch: int, n_kv: int, max_seq: int,
                  head_dim: int, device, dtype) -> List[Tuple[torch.Tensor,
                                                              torch.Tensor]]:
    return [(torch.zeros(batch, n_kv, max_seq, head_dim, device=device,
                         dtype=dtype),
             torch.zeros(batch, n_kv, max_seq, head_dim, device=device,
                         dtype=dtype))
            for _ in range(n_layer)]


def _sample(logits: torch.Tensor, greedy: bool, temperature: float,
            top_k: Optional[int]) -> torch.Tensor:
    """logits (B, V) -> next token ids (B,)."""
    if greedy:
        return logits.argmax(dim=-1)
    logits = logits.float() / max(temperature, 1e-5)
    if top_k is not None and top_k < logits.shape[-1]:
        kth = logits.topk(top_k, dim=-1).values[:, -1:]
        logits = logits.masked_fill(logits < kth, float("-inf"))
    probs = torch.softmax(logits, dim=-1)
    return torch.multinomial(probs, 1).squeeze(-1)


@torch.no_grad()
def _gpt2_decode_forward(model, idx: torch.Tensor, caches, pos0: int):
    """Forward ``idx`` (B, S_new) through the cached stack; returns the last
    position's logits (B, V)."""
    B, S = idx.shape
    pos = torch.arange(pos0, pos0 + S, device=idx.device)
    x = model.embed.wte(idx) + model.embed.wpe(pos)[None, :, :]
    x = x.transpose(0, 1).contiguous()          # (S, B, D)
    for blk, (kc, vc) in zip(model.blocks, caches):
        x = blk.decode_step(x, kc, vc, pos0)
    return model.head(x[-1:])[:, -1]            # ln_f + tied head, (B, V)


@torch.no_grad()
def _llama_decode_forward(model, idx: torch.Tensor, caches, pos0: int):
    B, S = idx.shape
    x = model.embed.tok(idx).transpose(0, 1).contiguous()
    for blk, (kc, vc) in zip(model.blocks, caches):
        x = blk.decode_step(x, kc, vc, pos0)
    return model.head(x[-1:])[:, -1]


@torch.no_grad()
def generate(model, idx: torch.Tensor, max_new_tokens: int,
             greedy: bool = True, temperature: float = 1.0,
             top_k: Optional[int] = None) -> torch.Tensor:
    """Autoregressive generation with KV caches.

    Args:
        model: a GPT2Model or LlamaModel (tp=1).
        idx: prompt token ids (B, S0).
        max_new_tokens: number of tokens to append.
    Returns (B, S0 + max_new_tokens) token ids.
    """
    assert get_tp_size() == 1, "generate() supports tp=1 (single device)"
    cfg = model.cfg
    B, S0 = idx.shape
    total = S0 + max_new_tokens
    assert total <= cfg.max_seq, (total, cfg.max_seq)
    dev = idx.device
    p = next(model.parameters())
    hd = cfg.dim // cfg.n_head
    n_kv = getattr(cfg, "n_kv_head", cfg.n_head)
    fwd = _llama_decode_forward if hasattr(model.embed, "tok") \
        else _gpt2_decode_forward
    caches = _alloc_caches(cfg.n_layer, B, n_kv, total, hd, dev, p.dtype)

    tokens = idx
    chunk, pos0 = idx, 0
    for _ in range(max_new_tokens):
        logits = fwd(model, chunk, caches, pos0)
        nxt = _sample(logits, greedy, temperature, top_k)
        tokens = torch.cat([tokens, nxt[:, None]], dim=1)
        pos0 += chunk.shape[1]
        chunk = nxt[:, None]
    return tokens
