"""Llama-3-style model family (RMSNorm, RoPE, SwiGLU, GQA).

BASELINE.json config 5: "Llama-3 8B hybrid intra-node ZeRO + sharded EMA,
bf16, 288 GB HBM sizing".  The reference has no model zoo; this is built on
the same TP primitives as GPT-2 (Col/Row parallel linears, in-tree HIP
RMSNorm + flash attention).

RoPE: cos/sin tables precomputed on host at init (guide Appendix B: on-device
trig turns memory-bound into VALU-bound), applied as a fused rotate-half in
bf16.  GQA: K/V heads repeated to match Q heads before the flash kernel
(kernel-native GQA is a planned kernel upgrade).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional

import torch
import torch.nn as nn

from ..ops import RMSNorm, flash_attention, rope_rotate_half, swiglu
from ..ops.gemm import linear as fast_linear
from ..parallel.tensor.tp_utils import (ColParallelLinear, RowParallelLinear,
                                        copy_to_tp_region,
                                        gather_from_sequence_parallel_region,
                                        get_tp_size, is_sequence_parallel,
                                        maybe_split_into_sequence_parallel,
                                        mark_sequence_parallel_params,
                                        set_sequence_parallel_attr)


@dataclass
class LlamaConfig:
    vocab_size: int = 128256
    n_layer: int = 32
    n_head: int = 32
    n_kv_head: int = 8
    dim: int = 4096
    ffn_dim: int = 14336
    max_seq: int = 8192
    rope_theta: float = 500000.0
    norm_eps: float = 1e-5
    causal: bool = True
    sequence_parallel: bool = True
    tie_weights: bool = False
    # vocab-sharded embedding/head/CE over TP (see models/gpt2.py); inert
    # at tp=1
    vocab_parallel: bool = True


def llama3_8b() -> LlamaConfig:
    return LlamaConfig()


def llama_tiny() -> LlamaConfig:
    return LlamaConfig(vocab_size=512, n_layer=2, n_head=4, n_kv_head=2,
                       dim=256, ffn_dim=512, max_seq=128)


class Rope(nn.Module):
    """Precomputed-table rotary embedding, rotate-half convention."""

    def __init__(self, head_dim: int, max_seq: int, theta: float,
                 device=None, dtype=None):
        super().__init__()
        inv = 1.0 / (theta ** (torch.arange(0, head_dim, 2).float() / head_dim))
        t = torch.arange(max_seq).float()
        freqs = torch.outer(t, inv).to(device)          # (S, hd/2)
        self.register_buffer("cos", freqs.cos(), persistent=False)
        self.register_buffer("sin", freqs.sin(), persistent=False)

    def forward(self, x: torch.Tensor, pos0: int = 0) -> torch.Tensor:
        # x (B, H, S, hd) — fused rotate-half kernel (ops/csrc/rope_swiglu.hip)
        return rope_rotate_half(x, self.cos, self.sin, pos0)


class LlamaAttention(nn.Module):
    """GQA attention with TP over heads (q heads and kv heads both divided by
    tp; n_kv_head % tp == 0 required)."""

    def __init__(self, cfg: LlamaConfig, device=None, dtype=None):
        super().__init__()
        tp = get_tp_size()
        assert cfg.n_head % tp == 0 and cfg.n_kv_head % tp == 0
        kw = {"device": device, "dtype": dtype}
        self.hd = cfg.dim // cfg.n_head
        self.nh_local = cfg.n_head // tp
        self.nkv_local = cfg.n_kv_head // tp
        self.causal = cfg.causal
        self.sequence_parallel = cfg.sequence_parallel and tp > 1
        q_out = cfg.dim
        kv_out = cfg.n_kv_head * self.hd
        self.wq = ColParallelLinear(cfg.dim, q_out, bias=False, **kw)
        self.wk = ColParallelLinear(cfg.dim, kv_out, bias=False, **kw)
        self.wv = ColParallelLinear(cfg.dim, kv_out, bias=False, **kw)
        self.wo = RowParallelLinear(cfg.dim, cfg.dim, bias=False,
                                    sequence_parallel=self.sequence_parallel,
                                    **kw)
        self.rope = Rope(self.hd, cfg.max_seq, cfg.rope_theta, **kw)

    def forward(self, x):
        if is_sequence_parallel(x):
            x = gather_from_sequence_parallel_region(x)
        else:
            x = copy_to_tp_region(x)
        S, B, _ = x.shape
        q = fast_linear(x, self.wq.weight)
        k = fast_linear(x, self.wk.weight)
        v = fast_linear(x, self.wv.weight)

        # zero-copy layouts: RoPE reads the permuted projection views in
        # place (strided kernel input); V goes to the flash kernel as a
        # strided view; the attention output is written straight into an
        # (S, B, H*hd) buffer for the out-projection GEMM.  The permuted
        # .contiguous() chain this replaces cost ~4 large copies per layer.
        def view4(t, nh):
            return t.reshape(S, B, nh, self.hd).permute(1, 2, 0, 3)

        q = self.rope(view4(q, self.nh_local))      # -> contiguous (B,H,S,D)
        k = self.rope(view4(k, self.nkv_local))
        v = view4(v, self.nkv_local)
        if q.is_cuda and q.dtype == torch.bfloat16 and self.hd in (64, 128):
            from ..ops import gqa_attention
            o = gqa_attention(q, k, v, causal=self.causal)
        else:
            o = flash_attention(q, k, v.contiguous(), causal=self.causal)
            o = o.permute(2, 0, 1, 3).reshape(S, B, self.nh_local * self.hd)
        return self.wo(o)

    @torch.no_grad()
    def decode_step(self, x: torch.Tensor, k_cache: torch.Tensor,
                    v_cache: torch.Tensor, pos0: int) -> torch.Tensor:
        """KV-cache inference step (tp=1 only; see inference/generate.py).
        RoPE is applied at the absolute positions ``pos0..pos0+S_new``; the
        cache holds rotated K, so cached entries are reused verbatim.  K/V
        heads are expanded to Q heads for eager SDPA (decode is GEMV-shaped,
        memory-bound — the flash kernel is the training-shape path)."""
        from ..parallel.tensor.attn import _cached_sdpa
        S, B, _ = x.shape
        q = fast_linear(x, self.wq.weight)
        k = fast_linear(x, self.wk.weight)
        v = fast_linear(x, self.wv.weight)

        def view4(t, nh):
            return t.reshape(S, B, nh, self.hd).permute(1, 2, 0, 3)

        q = self.rope(view4(q, self.nh_local), pos0)
        k = self.rope(view4(k, self.nkv_local), pos0)
        v = view4(v, self.nkv_local)
        k_cache[:, :, pos0:pos0 + S] = k
        v_cache[:, :, pos0:pos0 + S] = v
        rep = self.nh_local // self.nkv_local
        o = _cached_sdpa(q, k_cache.repeat_interleave(rep, dim=1),
                         v_cache.repeat_interleave(rep, dim=1), pos0,
                         self.causal)
        o = o.permute(2, 0, 1, 3).reshape(S, B, self.nh_local * self.hd)
        return self.wo(o)


class LlamaMlp(nn.Module):
    """SwiGLU: w2(silu(w1 x) * w3 x), TP col/col/row."""

    def __init__(self, cfg: LlamaConfig, device=None, dtype=None):
        super().__init__()
        kw = {"device": device, "dtype": dtype}
        self.sequence_parallel = cfg.sequence_parallel and get_tp_size() > 1
        self.w1 = ColParallelLinear(cfg.dim, cfg.ffn_dim, bias=False, **kw)
        self.w3 = ColParallelLinear(cfg.dim, cfg.ffn_dim, bias=False, **kw)
        self.w2 = RowParallelLinear(cfg.ffn_dim, cfg.dim, bias=False,
                                    sequence_parallel=self.sequence_parallel,
                                    **kw)

    def forward(self, x):
        if is_sequence_parallel(x):
            x = gather_from_sequence_parallel_region(x)
        else:
            x = copy_to_tp_region(x)
        return self.w2(swiglu(fast_linear(x, self.w1.weight),
                              fast_linear(x, self.w3.weight)))


class LlamaBlock(nn.Module):
    def __init__(self, cfg: LlamaConfig, device=None, dtype=None):
        super().__init__()
        kw = {"device": device, "dtype": dtype}
        self.sequence_parallel = cfg.sequence_parallel and get_tp_size() > 1
        self.attn_norm = RMSNorm(cfg.dim, cfg.norm_eps, **kw)
        self.attn = LlamaAttention(cfg, **kw)
        self.mlp_norm = RMSNorm(cfg.dim, cfg.norm_eps, **kw)
        self.mlp = LlamaMlp(cfg, **kw)
        if self.sequence_parallel:
            mark_sequence_parallel_params(self.attn_norm)
            mark_sequence_parallel_params(self.mlp_norm)

    def forward(self, x):
        if self.sequence_parallel:
            x = maybe_split_into_sequence_parallel(x)
        h = self.attn_norm(x)
        if self.sequence_parallel:
            set_sequence_parallel_attr(h)
        x = x + self.attn(h)
        h = self.mlp_norm(x)
        if self.sequence_parallel:
            set_sequence_parallel_attr(h)
        x = x + self.mlp(h)
        if self.sequence_parallel:
            set_sequence_parallel_attr(x)
        return x

    @torch.no_grad()
    def decode_step(self, x, k_cache, v_cache, pos0: int):
        x = x + self.attn.decode_step(self.attn_norm(x), k_cache, v_cache,
                                      pos0)
        return x + self.mlp(self.mlp_norm(x))


class LlamaEmbedding(nn.Module):
    def __init__(self, cfg: LlamaConfig, device=None, dtype=None):
        super().__init__()
        self.vocab_parallel = cfg.vocab_parallel and get_tp_size() > 1
        self.sequence_parallel = cfg.sequence_parallel
        if self.vocab_parallel:
            from ..parallel.tensor.vocab import VocabParallelEmbedding
            self.tok = VocabParallelEmbedding(cfg.vocab_size, cfg.dim,
                                              init_std=0.02, device=device,
                                              dtype=dtype)
        else:
            self.tok = nn.Embedding(cfg.vocab_size, cfg.dim, device=device,
                                    dtype=dtype)
            nn.init.normal_(self.tok.weight, std=0.02)

    def forward(self, idx):
        if self.vocab_parallel and self.sequence_parallel \
                and get_tp_size() > 1:
            # reduce-scatter straight into the (S/tp, B, D) SP layout
            x = self.tok(idx, sequence_parallel_out=True)
            set_sequence_parallel_attr(x)
            return x
        return self.tok(idx).transpose(0, 1).contiguous()  # (S, B, D)


class LlamaHead(nn.Module):
    def __init__(self, cfg: LlamaConfig, tok,
                 device=None, dtype=None):
        super().__init__()
        kw = {"device": device, "dtype": dtype}
        self.norm = RMSNorm(cfg.dim, cfg.norm_eps, **kw)
        self.vocab_parallel = cfg.vocab_parallel and get_tp_size() > 1
        if self.vocab_parallel:
            from ..parallel.tensor.vocab import VocabParallelHead
            tied = tok.weight if (cfg.tie_weights and tok is not None) \
                else None
            vh = VocabParallelHead(cfg.dim, cfg.vocab_size, weight=tied,
                                   init_std=0.02, **kw)
            self.vocab_start, self.vocab_end = vh.vocab_start, vh.vocab_end
            self.weight = tied if tied is not None else vh.weight
            if cfg.sequence_parallel:
                mark_sequence_parallel_params(self.norm)
        elif cfg.tie_weights and tok is not None:
            self.weight = tok.weight
        else:
            self.weight = nn.Parameter(
                torch.empty(cfg.vocab_size, cfg.dim, **kw))
            nn.init.normal_(self.weight, std=0.02)

    def forward(self, x):
        if self.vocab_parallel:
            sp_in = is_sequence_parallel(x)
            x = self.norm(x)
            if sp_in and get_tp_size() > 1:
                x = gather_from_sequence_parallel_region(
                    x, bwd_mode="reduce_scatter")
            else:
                x = copy_to_tp_region(x)
            return fast_linear(x, self.weight).transpose(0, 1)
        return fast_linear(self.norm(x), self.weight).transpose(0, 1)


class LlamaModel(nn.Module):
    def __init__(self, cfg: LlamaConfig, device=None, dtype=None):
        super().__init__()
        self.cfg = cfg
        kw = {"device": device, "dtype": dtype}
        self.embed = LlamaEmbedding(cfg, **kw)
        self.blocks = nn.ModuleList(
            [LlamaBlock(cfg, **kw) for _ in range(cfg.n_layer)])
        self.head = LlamaHead(cfg, self.embed.tok if cfg.tie_weights else None,
                              **kw)

    def forward(self, idx, labels: Optional[torch.Tensor] = None) -> dict:
        x = self.embed(idx)
        for blk in self.blocks:
            x = blk(x)
        if self.head.vocab_parallel:
            logits = self.head(x)   # LOCAL (B, S, V/tp), never gathered
            out = {"logits": logits}
            if labels is not None:
                from ..parallel.tensor.vocab import \
                    vocab_parallel_cross_entropy
                out["loss"] = vocab_parallel_cross_entropy(
                    logits.transpose(0, 1), labels.transpose(0, 1),
                    self.head.vocab_start, self.head.vocab_end)
            return out
        if is_sequence_parallel(x) and get_tp_size() > 1:
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

    def to_stage_layers(self) -> List[nn.Module]:
        return [self.embed, *self.blocks, self.head]

    def num_params(self) -> int:
        return sum(p.numel() for p in self.parameters())
