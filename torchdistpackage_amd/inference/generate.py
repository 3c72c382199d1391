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
def _gpt2_decode_forward(model, idx: torch.Tensor, caches, pos0: int,
                         all_logits: bool = False):
    """Forward ``idx`` (B, S_new) through the cached stack; returns the last
    position's logits (B, V), or every position's (B, S_new, V) with
    ``all_logits`` (speculative verification needs the whole chunk)."""
    B, S = idx.shape
    pos = torch.arange(pos0, pos0 + S, device=idx.device)
    x = model.embed.wte(idx) + model.embed.wpe(pos)[None, :, :]
    x = x.transpose(0, 1).contiguous()          # (S, B, D)
    for blk, (kc, vc) in zip(model.blocks, caches):
        x = blk.decode_step(x, kc, vc, pos0)
    if all_logits:
        return model.head(x)                    # (B, S, V)
    return model.head(x[-1:])[:, -1]            # ln_f + tied head, (B, V)


@torch.no_grad()
def _llama_decode_forward(model, idx: torch.Tensor, caches, pos0: int,
                          all_logits: bool = False):
    B, S = idx.shape
    x = model.embed.tok(idx).transpose(0, 1).contiguous()
    for blk, (kc, vc) in zip(model.blocks, caches):
        x = blk.decode_step(x, kc, vc, pos0)
    if all_logits:
        return model.head(x)
    return model.head(x[-1:])[:, -1]


class _GraphedDecoder:
    """Greedy single-token decode captured as ONE hipGraph.

    The eager decode step is launch-bound (~330 kernels for 24 layers,
    measured 5.3 ms/step on 1.3B); every shape here is static so the whole
    step — embedding, all blocks over the full-length cache with an
    additive visibility mask, head, argmax fed back into the input buffer —
    replays as a single graph launch (measured 2.5 ms/step, and 2.0 ms
    single-stream with the decode GEMV kernels).  Host work per token:
    advance the position tensor, open one mask slot, replay.

    Greedy only (the argmax lives inside the graph).  Subclasses bind the
    model family; usage::

        dec = GraphedGPT2Decoder(model, batch=B, max_seq=T)   # or Llama
        out = dec.generate(prompt, max_new_tokens=n)
    """

    def __init__(self, model, batch: int, max_seq: int, n_kv: int,
                 warmup: int = 3):
        assert get_tp_size() == 1
        assert torch.cuda.is_available()
        self.model = model
        cfg = model.cfg
        assert max_seq <= cfg.max_seq
        p = next(model.parameters())
        dev, dtype = p.device, p.dtype
        hd = cfg.dim // cfg.n_head
        self.max_seq = max_seq
        self.caches = _alloc_caches(cfg.n_layer, batch, n_kv, max_seq,
                                    hd, dev, dtype)
        self.in_tok = torch.zeros(batch, 1, dtype=torch.long, device=dev)
        self.pos_t = torch.zeros(1, dtype=torch.long, device=dev)
        self.mask = torch.full((1, 1, 1, max_seq), float("-inf"),
                               dtype=dtype, device=dev)
        self.pos = 0
        # capture with position 0 visible (an all-masked row would NaN);
        # the garbage this warmup writes into cache slot 0 is overwritten
        # by the first prefill()
        self.mask[..., :1] = 0
        self._graph = torch.cuda.CUDAGraph()
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(warmup):
                self._step_static()
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        with torch.cuda.graph(self._graph):
            self._step_static()

    def _step_static(self):
        raise NotImplementedError

    def _prefill_fwd(self, idx):
        raise NotImplementedError

    @torch.no_grad()
    def prefill(self, idx: torch.Tensor):
        """Run the prompt (B, S0) through the non-graph path into this
        decoder's caches and arm the first decode step."""
        B, S0 = idx.shape
        assert S0 < self.max_seq
        logits = self._prefill_fwd(idx)
        self.pos = S0
        self.mask.fill_(float("-inf"))
        self.mask[..., :S0 + 1] = 0
        self.pos_t.fill_(S0)
        self.in_tok.copy_(logits.argmax(-1, keepdim=True))

    @torch.no_grad()
    def step(self) -> torch.Tensor:
        """Emit one token per batch row: (B,) ids.  The returned token is
        the one the PREVIOUS forward predicted; this replay consumes it."""
        assert self.pos < self.max_seq, "decoder cache is full"
        tok = self.in_tok[:, 0].clone()
        self._graph.replay()
        self.pos += 1
        self.pos_t += 1
        if self.pos < self.max_seq:
            self.mask[..., self.pos] = 0
        return tok

    @torch.no_grad()
    def generate(self, idx: torch.Tensor,
                 max_new_tokens: int) -> torch.Tensor:
        assert idx.shape[1] + max_new_tokens <= self.max_seq
        self.prefill(idx)
        toks = [self.step() for _ in range(max_new_tokens)]
        return torch.cat([idx, torch.stack(toks, dim=1)], dim=1)


class GraphedGPT2Decoder(_GraphedDecoder):
    def __init__(self, model, batch: int, max_seq: int, warmup: int = 3):
        super().__init__(model, batch, max_seq, model.cfg.n_head, warmup)

    def _prefill_fwd(self, idx):
        return _gpt2_decode_forward(self.model, idx, self.caches, 0)

    @torch.no_grad()
    def _attn(self, attn, x):
        # x (1, B, D); cache write + full-length masked SDPA, all static
        from ..ops.gemm import linear as fast_linear
        B, D = x.shape[1], x.shape[2]
        hl = attn.n_head_local
        hd = D // attn.n_head
        qkv = fast_linear(x, attn.qkv.weight, attn.qkv.bias)
        q, k, v = qkv.split(hl * hd, dim=-1)

        def v4(t):
            return t.reshape(1, B, hl, hd).permute(1, 2, 0, 3)

        q, k, v = v4(q), v4(k), v4(v)
        kc, vc = self._cur_cache
        kc.index_copy_(2, self.pos_t, k)
        vc.index_copy_(2, self.pos_t, v)
        o = torch.nn.functional.scaled_dot_product_attention(
            q, kc, vc, attn_mask=self.mask)
        o = o.permute(2, 0, 1, 3).reshape(1, B, hl * hd)
        return attn.proj(o)

    @torch.no_grad()
    def _step_static(self):
        m = self.model
        x = m.embed.wte(self.in_tok) + m.embed.wpe(self.pos_t)[None, :, :]
        x = x.transpose(0, 1)                    # (1, B, D)
        for blk, cache in zip(m.blocks, self.caches):
            self._cur_cache = cache
            h = blk.ln_1(x)
            x = x + self._attn(blk.attn, h)
            x = x + blk.mlp(blk.ln_2(x))
        logits = m.head(x)[:, -1]                # (B, V)
        self._logits = logits
        self.in_tok.copy_(logits.argmax(-1, keepdim=True))


class GraphedLlamaDecoder(_GraphedDecoder):
    """Llama-family graphed decode: RoPE reads its position from the
    device tensor (ops rope_apply_pos — the host-scalar variant would bake
    one position into the graph), K/V heads expand to Q heads for SDPA."""

    def __init__(self, model, batch: int, max_seq: int, warmup: int = 3):
        super().__init__(model, batch, max_seq, model.cfg.n_kv_head, warmup)

    def _prefill_fwd(self, idx):
        return _llama_decode_forward(self.model, idx, self.caches, 0)

    @torch.no_grad()
    def _attn(self, attn, x):
        from ..ops.gemm import linear as fast_linear
        B, D = x.shape[1], x.shape[2]
        hd = attn.hd
        q = fast_linear(x, attn.wq.weight)
        k = fast_linear(x, attn.wk.weight)
        v = fast_linear(x, attn.wv.weight)

        def v4(t, nh):
            return t.reshape(1, B, nh, hd).permute(1, 2, 0, 3)

        q = attn.rope(v4(q, attn.nh_local), self.pos_t)
        k = attn.rope(v4(k, attn.nkv_local), self.pos_t)
        v = v4(v, attn.nkv_local)
        kc, vc = self._cur_cache
        kc.index_copy_(2, self.pos_t, k)
        vc.index_copy_(2, self.pos_t, v)
        rep = attn.nh_local // attn.nkv_local
        o = torch.nn.functional.scaled_dot_product_attention(
            q, kc.repeat_interleave(rep, dim=1),
            vc.repeat_interleave(rep, dim=1), attn_mask=self.mask)
        o = o.permute(2, 0, 1, 3).reshape(1, B, attn.nh_local * hd)
        return attn.wo(o)

    @torch.no_grad()
    def _step_static(self):
        m = self.model
        x = m.embed.tok(self.in_tok).transpose(0, 1)   # (1, B, D)
        for blk, cache in zip(m.blocks, self.caches):
            self._cur_cache = cache
            x = x + self._attn(blk.attn, blk.attn_norm(x))
            x = x + blk.mlp(blk.mlp_norm(x))
        logits = m.head(x)[:, -1]
        self._logits = logits
        self.in_tok.copy_(logits.argmax(-1, keepdim=True))


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


@torch.no_grad()
def speculative_generate(target, draft, idx: torch.Tensor,
                         max_new_tokens: int, k: int = 4) -> torch.Tensor:
    """Greedy speculative decoding: ``draft`` proposes ``k`` tokens per
    round, ``target`` verifies the whole chunk in ONE forward.  Output is
    EXACTLY the target's greedy generation — the draft only changes how
    many target forwards it takes (verified by the differential test).

    Cache bookkeeping: rejected proposals leave stale entries past the
    committed length in both caches; they are overwritten sequentially
    before any later position can attend to them, so a rollback is just a
    smaller next ``pos0``.  B=1 (speculative decode is per-request —
    batched rows would accept different lengths each round).
    """
    assert get_tp_size() == 1
    assert idx.shape[0] == 1, "speculative decode is per-request (B=1)"
    t_fwd = _llama_decode_forward if hasattr(target.embed, "tok") \
        else _gpt2_decode_forward
    d_fwd = _llama_decode_forward if hasattr(draft.embed, "tok") \
        else _gpt2_decode_forward
    S0 = idx.shape[1]
    cap = S0 + max_new_tokens + k + 1
    assert cap <= target.cfg.max_seq and cap <= draft.cfg.max_seq
    dev = idx.device

    def caches_of(m):
        cfg = m.cfg
        return _alloc_caches(cfg.n_layer, 1,
                             getattr(cfg, "n_kv_head", cfg.n_head), cap,
                             cfg.dim // cfg.n_head, dev,
                             next(m.parameters()).dtype)

    t_caches, d_caches = caches_of(target), caches_of(draft)
    lt = t_fwd(target, idx, t_caches, 0)
    d_fwd(draft, idx, d_caches, 0)
    tokens = torch.cat([idx, lt.argmax(-1)[:, None]], dim=1)
    produced = 1
    while produced < max_new_tokens:
        kk = min(k, max_new_tokens - produced)
        base = tokens.shape[1] - 1        # cache-valid length of both models
        # ---- draft proposes kk tokens after the last committed token
        props = []
        cur = tokens[:, -1:]
        for j in range(kk):
            dl = d_fwd(draft, cur, d_caches, base + j)
            cur = dl.argmax(-1)[:, None]
            props.append(cur)
        # ---- target verifies [last_committed, props[:-1]] in one chunk
        chunk = torch.cat([tokens[:, -1:]] + props[:-1], dim=1)  # (1, kk)
        tl = t_fwd(target, chunk, t_caches, base, all_logits=True)
        truth = tl.argmax(-1)             # (1, kk): token after each prefix
        a = 0
        while a < kk - 1 and bool(truth[0, a] == props[a][0, 0]):
            a += 1
        # accept props[0..a-1] (they matched) plus the target's token at a
        accepted = props[:a] + [truth[:, a:a + 1]]
        tokens = torch.cat([tokens] + accepted, dim=1)
        produced += a + 1
    return tokens[:, :S0 + max_new_tokens]
