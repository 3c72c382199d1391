"""Hand-written gfx950 HIP ops for the transformer hot path.

These replace the implicit torch-op hot path the reference relies on
(layernorm, naive SDPA, GELU, Adam step, EMA update,
grad-norm/clip) with CDNA4 kernels (MFMA for attention, LDS-tiled /
vectorized bf16 for the memory-bound ops).

Dispatch policy:
- CUDA (= HIP/MI355X) tensors: the in-tree extension ``_tdpa_hip`` MUST be
  loaded — a missing extension raises instead of silently falling back to
  eager torch (the silent-fallback trap the build rules call out).
- CPU tensors: plain-torch reference implementations (used by the gloo CI
  tests, and as the numerics oracle for the GPU kernels).
"""

from __future__ import annotations

import math
import os
from typing import Optional

import torch

_EXT = None
_EXT_ERR: Optional[str] = None


def _load_extension():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    import glob
    import importlib.util
    here = os.path.dirname(os.path.abspath(__file__))
    cands = glob.glob(os.path.join(here, "_tdpa_hip*.so"))
    if not cands:
        _EXT_ERR = (f"in-tree HIP extension not found under {here}; "
                    "run `python -m torchdistpackage_amd.ops.build` "
                    "(or __graft_entry__.build())")
        return None
    spec = importlib.util.spec_from_file_location("_tdpa_hip", cands[0])
    mod = importlib.util.module_from_spec(spec)
    try:
        spec.loader.exec_module(mod)
    except Exception as e:  # loud, not silent
        _EXT_ERR = f"failed to load {cands[0]}: {e}"
        return None
    _EXT = mod
    return _EXT


def ext(required_for: str = "op"):
    """Return the extension module; raise if a GPU op needs it and it's
    missing (no silent eager fallback on GPU)."""
    m = _load_extension()
    if m is None:
        raise RuntimeError(
            f"HIP extension required for {required_for} on GPU but "
            f"unavailable: {_EXT_ERR}")
    return m


def extension_available() -> bool:
    return _load_extension() is not None


# ---------------------------------------------------------------------------
# RMSNorm
# ---------------------------------------------------------------------------

class _RMSNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        if x.is_cuda:
            y, rstd = ext("rms_norm").rmsnorm_fwd(x.contiguous(), weight, eps)
        else:
            xf = x.float()
            rstd = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
            y = (xf * rstd * weight.float()).to(x.dtype)
            rstd = rstd.squeeze(-1)
        ctx.save_for_backward(x, weight, rstd)
        ctx.eps = eps
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, rstd = ctx.saved_tensors
        if x.is_cuda:
            dx, dw = ext("rms_norm").rmsnorm_bwd(
                dy.contiguous(), x.contiguous(), weight, rstd)
        else:
            xf, dyf, wf = x.float(), dy.float(), weight.float()
            r = rstd.unsqueeze(-1)
            xhat = xf * r
            wdy = dyf * wf
            # dx = r * (wdy - xhat * mean(wdy * xhat))
            m = (wdy * xhat).mean(-1, keepdim=True)
            dx = (r * (wdy - xhat * m)).to(x.dtype)
            dw = (dyf * xhat).reshape(-1, x.shape[-1]).sum(0).to(weight.dtype)
        return dx, dw, None


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6):
    return _RMSNormFn.apply(x, weight, eps)


class RMSNorm(torch.nn.Module):
    def __init__(self, dim: int, eps: float = 1e-6, device=None, dtype=None):
        super().__init__()
        self.eps = eps
        self.weight = torch.nn.Parameter(
            torch.ones(dim, device=device, dtype=dtype))

    def forward(self, x):
        return rms_norm(x, self.weight, self.eps)


# ---------------------------------------------------------------------------
# LayerNorm
# ---------------------------------------------------------------------------

class _LayerNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        if x.is_cuda:
            y, mean, rstd = ext("layer_norm").layernorm_fwd(
                x.contiguous(), weight, bias, eps)
        else:
            xf = x.float()
            mean = xf.mean(-1, keepdim=True)
            var = xf.var(-1, unbiased=False, keepdim=True)
            rstd = torch.rsqrt(var + eps)
            y = ((xf - mean) * rstd * weight.float() + bias.float()).to(x.dtype)
            mean, rstd = mean.squeeze(-1), rstd.squeeze(-1)
        ctx.save_for_backward(x, weight, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, mean, rstd = ctx.saved_tensors
        if x.is_cuda:
            dx, dw, db = ext("layer_norm").layernorm_bwd(
                dy.contiguous(), x.contiguous(), weight, mean, rstd)
        else:
            xf, dyf, wf = x.float(), dy.float(), weight.float()
            mu, r = mean.unsqueeze(-1), rstd.unsqueeze(-1)
            xhat = (xf - mu) * r
            wdy = dyf * wf
            m1 = wdy.mean(-1, keepdim=True)
            m2 = (wdy * xhat).mean(-1, keepdim=True)
            dx = (r * (wdy - m1 - xhat * m2)).to(x.dtype)
            dw = (dyf * xhat).reshape(-1, x.shape[-1]).sum(0).to(weight.dtype)
            db = dyf.reshape(-1, x.shape[-1]).sum(0).to(weight.dtype)
        return dx, dw, db, None


def layer_norm(x, weight, bias, eps: float = 1e-5):
    return _LayerNormFn.apply(x, weight, bias, eps)


class LayerNorm(torch.nn.Module):
    def __init__(self, dim: int, eps: float = 1e-5, device=None, dtype=None):
        super().__init__()
        self.eps = eps
        kw = {"device": device, "dtype": dtype}
        self.weight = torch.nn.Parameter(torch.ones(dim, **kw))
        self.bias = torch.nn.Parameter(torch.zeros(dim, **kw))

    def forward(self, x):
        return layer_norm(x, self.weight, self.bias, self.eps)


# ---------------------------------------------------------------------------
# fused bias + GELU (tanh approximation, GPT-2 convention)
# ---------------------------------------------------------------------------

_GELU_C = math.sqrt(2.0 / math.pi)


def _gelu_tanh(u):
    return 0.5 * u * (1.0 + torch.tanh(_GELU_C * (u + 0.044715 * u.pow(3))))


class _BiasGeluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, bias):
        ctx.has_bias = bias is not None
        if x.is_cuda:
            y = ext("bias_gelu").bias_gelu_fwd(
                x.contiguous(),
                bias if bias is not None else torch.empty(
                    0, dtype=x.dtype, device=x.device))
            ctx.save_for_backward(x, bias if bias is not None else
                                  torch.empty(0, dtype=x.dtype, device=x.device))
        else:
            u = x.float() + (bias.float() if bias is not None else 0.0)
            y = _gelu_tanh(u).to(x.dtype)
            ctx.save_for_backward(x, bias if bias is not None else
                                  torch.empty(0, dtype=x.dtype))
        return y

    @staticmethod
    def backward(ctx, dy):
        x, bias = ctx.saved_tensors
        if x.is_cuda:
            dx = ext("bias_gelu").bias_gelu_bwd(dy.contiguous(),
                                                x.contiguous(), bias)
        else:
            u = x.float() + (bias.float() if ctx.has_bias else 0.0)
            t = torch.tanh(_GELU_C * (u + 0.044715 * u.pow(3)))
            du = 0.5 * (1 + t) + 0.5 * u * (1 - t * t) * _GELU_C * \
                (1 + 3 * 0.044715 * u.pow(2))
            dx = (dy.float() * du).to(x.dtype)
        dbias = None
        if ctx.has_bias:
            dbias = dx.reshape(-1, x.shape[-1]).sum(0).to(x.dtype)
        return dx, dbias


def bias_gelu(x, bias=None):
    return _BiasGeluFn.apply(x, bias)


# ---------------------------------------------------------------------------
# flash attention (bf16, causal/full, head_dim 64/128)
# ---------------------------------------------------------------------------

class _FlashAttentionFn(torch.autograd.Function):
    """q: (B, H, S, D); k, v: (B, H_kv, S, D) with H % H_kv == 0 (GQA).
    GPU path = in-tree HIP flash kernels (blockwise online-softmax, spec:
    reference explore/flash-attn/tile_attn.py:100-212); CPU path = exact
    math attention in fp32 (the numerics oracle)."""

    @staticmethod
    def forward(ctx, q, k, v, causal, scale):
        if scale is None:
            scale = 1.0 / math.sqrt(q.shape[-1])
        rep = q.shape[1] // k.shape[1]
        if q.is_cuda:
            o = torch.empty(q.shape, dtype=q.dtype, device=q.device)
            o, lse = ext("flash_attention").attn_fwd(
                q, k, v, o, bool(causal), float(scale))
        else:
            kf = k.float() if rep == 1 else \
                k.float().repeat_interleave(rep, 1)
            vf = v.float() if rep == 1 else \
                v.float().repeat_interleave(rep, 1)
            qf = q.float()
            s = torch.matmul(qf, kf.transpose(-1, -2)) * scale
            if causal:
                S = q.shape[-2]
                mask = torch.ones(S, k.shape[-2], dtype=torch.bool,
                                  device=q.device).tril_()
                s = s.masked_fill(~mask, float("-inf"))
            lse = torch.logsumexp(s, dim=-1)
            p = torch.softmax(s, dim=-1)
            o = torch.matmul(p, vf).to(q.dtype)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.causal = causal
        ctx.scale = scale
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        H, Hkv = q.shape[1], k.shape[1]
        rep = H // Hkv
        if q.is_cuda:
            dq = torch.empty(q.shape, dtype=q.dtype, device=q.device)
            # GQA: kernels write per-q-head dk/dv partials; summed below
            dk = torch.empty(q.shape, dtype=k.dtype, device=k.device)
            dv = torch.empty(q.shape, dtype=v.dtype, device=v.device)
            dq, dk, dv = ext("flash_attention").attn_bwd(
                do.contiguous(), q, k, v, o, lse, dq, dk, dv,
                ctx.causal, ctx.scale)
            if rep > 1:
                B, _, S, D = q.shape
                dk = dk.view(B, Hkv, rep, S, D).sum(2).to(k.dtype)
                dv = dv.view(B, Hkv, rep, S, D).sum(2).to(v.dtype)
        else:
            qf = q.float()
            kf = k.float() if rep == 1 else \
                k.float().repeat_interleave(rep, 1)
            vf = v.float() if rep == 1 else \
                v.float().repeat_interleave(rep, 1)
            dof = do.float()
            s = torch.matmul(qf, kf.transpose(-1, -2)) * ctx.scale
            if ctx.causal:
                S = q.shape[-2]
                mask = torch.ones(S, k.shape[-2], dtype=torch.bool,
                                  device=q.device).tril_()
                s = s.masked_fill(~mask, float("-inf"))
            p = torch.softmax(s, dim=-1)
            dv = torch.matmul(p.transpose(-1, -2), dof)
            dp = torch.matmul(dof, vf.transpose(-1, -2))
            d = (dp * p).sum(-1, keepdim=True)
            ds = p * (dp - d) * ctx.scale
            dq = torch.matmul(ds, kf)
            dk = torch.matmul(ds.transpose(-1, -2), qf)
            if rep > 1:
                B, _, S, D = q.shape
                dk = dk.view(B, Hkv, rep, S, D).sum(2)
                dv = dv.view(B, Hkv, rep, S, D).sum(2)
            dq, dk, dv = dq.to(q.dtype), dk.to(k.dtype), dv.to(v.dtype)
        return dq, dk, dv, None, None


_warned_hd = set()


def flash_attention(q, k, v, causal: bool = True, scale: float = None):
    """Flash attention; the in-tree HIP kernels cover head_dim 64 and 128.
    Other head dims (80/96/...) go through an EXPLICIT composite path
    (torch SDPA with fp32 math) — correct but unfused; round-1 hard-failed
    these.  The fallback is announced once so a GPU
    run can never silently lose the kernels for supported dims."""
    D = q.shape[-1]
    if q.is_cuda and D not in (64, 128):
        if D not in _warned_hd:
            _warned_hd.add(D)
            print(f"[tdpa.ops] flash_attention: head_dim {D} has no HIP "
                  f"kernel (64/128 do); using composite SDPA for it")
        rep = q.shape[1] // k.shape[1]
        kk = k if rep == 1 else k.repeat_interleave(rep, 1)
        vv = v if rep == 1 else v.repeat_interleave(rep, 1)
        if scale is None:
            scale = 1.0 / math.sqrt(D)
        return torch.nn.functional.scaled_dot_product_attention(
            q, kk, vv, is_causal=causal, scale=scale)
    return _FlashAttentionFn.apply(q, k, v, causal, scale)


# ---------------------------------------------------------------------------
# fused optimizer / EMA / grad utilities (flat-tensor kernels)
# ---------------------------------------------------------------------------

def fused_adamw_(param: torch.Tensor, grad: torch.Tensor,
                 exp_avg: torch.Tensor, exp_avg_sq: torch.Tensor,
                 step: int, lr: float, beta1: float, beta2: float,
                 eps: float, weight_decay: float):
    """AdamW update on (flat) fp32 tensors, one fused kernel pass on GPU."""
    if param.is_cuda:
        ext("fused_adamw").adamw_step(param, grad, exp_avg, exp_avg_sq,
                                      step, lr, beta1, beta2, eps,
                                      weight_decay)
        return
    bc1 = 1 - beta1 ** step
    bc2 = 1 - beta2 ** step
    param.mul_(1 - lr * weight_decay)
    exp_avg.mul_(beta1).add_(grad, alpha=1 - beta1)
    exp_avg_sq.mul_(beta2).addcmul_(grad, grad, value=1 - beta2)
    denom = (exp_avg_sq / bc2).sqrt_().add_(eps)
    param.addcdiv_(exp_avg, denom, value=-lr / bc1)


def ema_update_(ema: torch.Tensor, param: torch.Tensor, decay: float):
    """ema = decay*ema + (1-decay)*param, fused."""
    if ema.is_cuda:
        ext("ema_update").ema_update(ema, param, decay)
        return
    ema.lerp_(param.to(ema.dtype), 1.0 - decay)


def l2norm_sq(t: torch.Tensor) -> torch.Tensor:
    """Sum of squares (fp32 accumulate) of one (flat) tensor."""
    if t.is_cuda:
        return ext("l2norm").l2norm_sq(t)
    return t.float().pow(2).sum()


def scale_(t: torch.Tensor, scale: float):
    if t.is_cuda:
        ext("scale").scale_inplace(t, scale)
        return
    t.mul_(scale)


class _FusedQKVAttentionFn(torch.autograd.Function):
    """Flash attention straight on the fused qkv projection output.

    qkv: (S, B, 3*H*hd) with per-rank layout [q | k | v]; returns (S, B, H*hd).
    The kernels read/write STRIDED (B,H,S,D) views into the fused buffers, so
    there is no permute-contiguous copy anywhere on this path (profiling of
    the copy-based path showed ~5%% of step time in transpose copies).
    CPU fallback delegates to the reference math in _FlashAttentionFn.
    """

    @staticmethod
    def forward(ctx, qkv, n_head, causal, scale):
        S, B, three_d = qkv.shape
        d_local = three_d // 3
        hd = d_local // n_head
        if scale is None:
            scale = 1.0 / math.sqrt(hd)

        def view4(t, off):
            return t.narrow(-1, off, d_local) \
                .view(S, B, n_head, hd).permute(1, 2, 0, 3)

        q, k, v = view4(qkv, 0), view4(qkv, d_local), view4(qkv, 2 * d_local)
        if qkv.is_cuda:
            o_buf = torch.empty(S, B, d_local, dtype=qkv.dtype,
                                device=qkv.device)
            o4 = o_buf.view(S, B, n_head, hd).permute(1, 2, 0, 3)
            _, lse = ext("flash_attention").attn_fwd(
                q, k, v, o4, bool(causal), float(scale))
        else:
            qf = q.contiguous().float()
            kf = k.contiguous().float()
            vf = v.contiguous().float()
            sc = torch.matmul(qf, kf.transpose(-1, -2)) * scale
            if causal:
                mask = torch.ones(S, S, dtype=torch.bool).tril_()
                sc = sc.masked_fill(~mask, float("-inf"))
            lse = torch.logsumexp(sc, dim=-1)
            o4 = torch.matmul(torch.softmax(sc, -1), vf).to(qkv.dtype)
            o_buf = o4.permute(2, 0, 1, 3).reshape(S, B, d_local)
        ctx.save_for_backward(qkv, o_buf, lse)
        ctx.meta = (n_head, causal, scale)
        return o_buf

    @staticmethod
    def backward(ctx, do):
        qkv, o_buf, lse = ctx.saved_tensors
        n_head, causal, scale = ctx.meta
        S, B, three_d = qkv.shape
        d_local = three_d // 3
        hd = d_local // n_head

        def view4(t, off, width):
            return t.narrow(-1, off, width) \
                .view(S, B, n_head, hd).permute(1, 2, 0, 3)

        q = view4(qkv, 0, d_local)
        k = view4(qkv, d_local, d_local)
        v = view4(qkv, 2 * d_local, d_local)
        o4 = o_buf.view(S, B, n_head, hd).permute(1, 2, 0, 3)
        do = do.contiguous()
        do4 = do.view(S, B, n_head, hd).permute(1, 2, 0, 3)
        if qkv.is_cuda:
            dqkv = torch.empty_like(qkv)
            dq = view4(dqkv, 0, d_local)
            dk = view4(dqkv, d_local, d_local)
            dv = view4(dqkv, 2 * d_local, d_local)
            ext("flash_attention").attn_bwd(
                do4, q, k, v, o4, lse, dq, dk, dv, causal, scale)
        else:
            qc, kc, vc = q.contiguous(), k.contiguous(), v.contiguous()
            with torch.enable_grad():  # Function.backward runs under no_grad
                qf = qc.float().requires_grad_(True)
                kf = kc.float().requires_grad_(True)
                vf = vc.float().requires_grad_(True)
                s = torch.matmul(qf, kf.transpose(-1, -2)) * scale
                if causal:
                    Sq = q.shape[-2]
                    mask = torch.ones(Sq, Sq, dtype=torch.bool).tril_()
                    s = s.masked_fill(~mask, float("-inf"))
                ref = torch.matmul(torch.softmax(s, -1), vf)
                ref.backward(do4.float())
            dqkv = torch.cat([
                g.to(qkv.dtype).permute(2, 0, 1, 3).reshape(S, B, d_local)
                for g in (qf.grad, kf.grad, vf.grad)], dim=-1)
        return dqkv, None, None, None



def fused_qkv_attention(qkv, n_head: int, causal: bool = True,
                        scale: float = None):
    return _FusedQKVAttentionFn.apply(qkv, n_head, causal, scale)


class _GqaAttentionFn(torch.autograd.Function):
    """Flash attention for the Llama path with zero-copy layouts.

    q (B,H,S,D) contiguous (RoPE output); k (B,Hkv,S,D) contiguous; v may be
    a STRIDED view of the (S,B,Hkv*hd) projection output.  Returns
    (S, B, H*hd) — the kernels write the output through a strided view, so
    no permute-contiguous copy feeds the out-projection GEMM.  dq is
    likewise written straight into an (S,B,H*hd)-backed view; GQA dk/dv
    partials (q-head width) are summed to kv width.
    """

    @staticmethod
    def forward(ctx, q, k, v, causal, scale):
        B, H, S, D = q.shape
        o_buf = torch.empty(S, B, H * D, dtype=q.dtype, device=q.device)
        o4 = o_buf.view(S, B, H, D).permute(1, 2, 0, 3)
        _, lse = ext("flash_attention").attn_fwd(
            q, k, v, o4, bool(causal), float(scale))
        ctx.save_for_backward(q, k, v, o_buf, lse)
        ctx.meta = (causal, scale)
        return o_buf

    @staticmethod
    def backward(ctx, do):
        q, k, v, o_buf, lse = ctx.saved_tensors
        causal, scale = ctx.meta
        B, H, S, D = q.shape
        Hkv = k.shape[1]
        rep = H // Hkv
        o4 = o_buf.view(S, B, H, D).permute(1, 2, 0, 3)
        do = do.contiguous()
        do4 = do.view(S, B, H, D).permute(1, 2, 0, 3)
        dq_buf = torch.empty(S, B, H * D, dtype=q.dtype, device=q.device)
        dq = dq_buf.view(S, B, H, D).permute(1, 2, 0, 3)
        dk = torch.empty(B, H, S, D, dtype=k.dtype, device=k.device)
        dv = torch.empty(B, H, S, D, dtype=v.dtype, device=v.device)
        ext("flash_attention").attn_bwd(do4, q, k, v, o4, lse, dq, dk, dv,
                                        causal, scale)
        if rep > 1:
            dk = dk.view(B, Hkv, rep, S, D).sum(2)
            dv = dv.view(B, Hkv, rep, S, D).sum(2)
        return dq, dk, dv, None, None


def gqa_attention(q, k, v, causal: bool = True, scale: float = None):
    """Zero-copy GQA flash attention (GPU bf16 path; see _GqaAttentionFn).
    Falls back to :func:`flash_attention` off-GPU / off-bf16."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if q.is_cuda and q.dtype == torch.bfloat16 and q.shape[-1] in (64, 128):
        return _GqaAttentionFn.apply(q, k, v, causal, scale)
    o = flash_attention(q, k, v, causal=causal, scale=scale)
    B, H, S, D = q.shape
    return o.permute(2, 0, 1, 3).reshape(S, B, H * D)


class _CrossEntropyFn(torch.autograd.Function):
    """Fused mean cross-entropy on bf16 logits (one online-LSE pass fwd, one
    dlogits pass bwd) — replaces logits.float()+F.cross_entropy which
    materializes fp32 logits.  CPU fallback is exact F.cross_entropy."""

    @staticmethod
    def forward(ctx, logits2d, targets):
        loss_vec, lse = ext("cross_entropy").ce_fwd(logits2d, targets)
        ctx.save_for_backward(logits2d, targets, lse)
        return loss_vec.mean()

    @staticmethod
    def backward(ctx, g):
        logits2d, targets, lse = ctx.saved_tensors
        dlogits = ext("cross_entropy").ce_bwd(
            logits2d, targets, lse, g.reshape(1).float().contiguous())
        return dlogits, None


def cross_entropy_loss(logits: torch.Tensor, targets: torch.Tensor):
    """Mean CE over flattened (N, V) logits; fused bf16 kernel on GPU."""
    l2 = logits.reshape(-1, logits.size(-1))
    t = targets.reshape(-1)
    if l2.is_cuda and l2.dtype == torch.bfloat16:
        return _CrossEntropyFn.apply(l2.contiguous(), t)
    import torch.nn.functional as F
    return F.cross_entropy(l2.float(), t)


# ---------------------------------------------------------------------------
# RoPE (rotate-half) and SwiGLU — fused Llama hot-path ops
# ---------------------------------------------------------------------------

class _BatchedBiasGeluFn(torch.autograd.Function):
    """gelu(x + bias[e]) over (E, N, H) with per-expert bias — the MoE
    batched-expert epilogue (hipBLASLt baddbmm faulted on a stride-0
    broadcast batch bias; bmm + this fused kernel replaces it)."""

    @staticmethod
    def forward(ctx, x, bias):
        ctx.save_for_backward(x, bias)
        if x.is_cuda and x.dtype == torch.bfloat16:
            return ext("bgelu_b").bgelu_b_fwd(x.contiguous(),
                                              bias.contiguous())
        u = x.float() + bias.float().unsqueeze(1)
        return _gelu_tanh(u).to(x.dtype)

    @staticmethod
    def backward(ctx, dy):
        x, bias = ctx.saved_tensors
        if x.is_cuda and x.dtype == torch.bfloat16:
            dx = ext("bgelu_b").bgelu_b_bwd(dy.contiguous(), x.contiguous(),
                                            bias.contiguous())
        else:
            u = x.float() + bias.float().unsqueeze(1)
            t = torch.tanh(_GELU_C * (u + 0.044715 * u.pow(3)))
            du = 0.5 * (1 + t) + 0.5 * u * (1 - t * t) * _GELU_C * \
                (1 + 3 * 0.044715 * u.pow(2))
            dx = (dy.float() * du).to(x.dtype)
        dbias = dx.float().sum(1).to(bias.dtype)
        return dx, dbias


def batched_bias_gelu(x, bias):
    return _BatchedBiasGeluFn.apply(x, bias)


class _RopeFn(torch.autograd.Function):
    """Rotate-half RoPE over (B, H, S, D) with fp32 (S, D/2) tables."""

    @staticmethod
    def forward(ctx, x, cos, sin, pos0):
        ctx.save_for_backward(cos, sin)
        ctx.pos0 = pos0
        if x.is_cuda and x.dtype == torch.bfloat16:
            # strided input is read in place (head dim must be contiguous);
            # output is contiguous (B, H, S, D)
            return ext("rope").rope_apply(x, cos, sin, pos0, True)
        hd = x.shape[-1]
        S = x.shape[-2]
        c = cos[pos0:pos0 + S].to(x.dtype)
        s = sin[pos0:pos0 + S].to(x.dtype)
        x1, x2 = x[..., :hd // 2], x[..., hd // 2:]
        return torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], dim=-1)

    @staticmethod
    def backward(ctx, dy):
        cos, sin = ctx.saved_tensors
        if dy.is_cuda and dy.dtype == torch.bfloat16:
            if dy.stride(-1) != 1:
                dy = dy.contiguous()
            return (ext("rope").rope_apply(dy, cos, sin,
                                           ctx.pos0, False),
                    None, None, None)
        hd = dy.shape[-1]
        S = dy.shape[-2]
        c = cos[ctx.pos0:ctx.pos0 + S].to(dy.dtype)
        s = sin[ctx.pos0:ctx.pos0 + S].to(dy.dtype)
        d1, d2 = dy[..., :hd // 2], dy[..., hd // 2:]
        return (torch.cat([d1 * c + d2 * s, d2 * c - d1 * s], dim=-1),
                None, None, None)


def rope_rotate_half(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
                     pos0=0) -> torch.Tensor:
    """Fused rotate-half rotary embedding (one read+write pass; the eager
    form is 2 cats + 4 muls per call, models/llama.py r01).

    ``pos0`` may be a 1-element long DEVICE tensor (inference only): the
    kernel reads the position at run time, which keeps the op
    hipGraph-capturable (inference/generate.py's graphed Llama decoder
    advances the tensor in place between replays)."""
    if torch.is_tensor(pos0):
        assert not torch.is_grad_enabled(), \
            "tensor pos0 is the no-grad decode path"
        if x.is_cuda and x.dtype == torch.bfloat16:
            return ext("rope").rope_apply_pos(x, cos, sin, pos0, True)
        return _RopeFn.apply(x, cos, sin, int(pos0.item()))
    return _RopeFn.apply(x, cos, sin, pos0)


class _SwiGluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, a, b):
        ctx.save_for_backward(a, b)
        if a.is_cuda and a.dtype == torch.bfloat16:
            return ext("swiglu").swiglu_fwd(a.contiguous(), b.contiguous())
        af = a.float()
        return (af * torch.sigmoid(af) * b.float()).to(a.dtype)

    @staticmethod
    def backward(ctx, dy):
        a, b = ctx.saved_tensors
        if a.is_cuda and a.dtype == torch.bfloat16:
            da, db = ext("swiglu").swiglu_bwd(dy.contiguous(),
                                              a.contiguous(), b.contiguous())
            return da, db
        af, bf, dyf = a.float(), b.float(), dy.float()
        sig = torch.sigmoid(af)
        silu = af * sig
        da = dyf * bf * (sig + silu * (1 - sig))
        db = dyf * silu
        return da.to(a.dtype), db.to(b.dtype)


def swiglu(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """silu(a) * b, fused (backward recomputes sigmoid — nothing extra
    saved beyond a and b)."""
    return _SwiGluFn.apply(a, b)
