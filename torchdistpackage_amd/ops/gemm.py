"""Dispatch layer for the hand-written CDNA4 GEMM (csrc/gemm.hip).

``linear(x, w, bias)`` is a drop-in for ``F.linear`` on the transformer hot
path.  Eligible (GPU, bf16, 256-aligned) shapes route through ONE autograd
Function whose forward/dgrad/wgrad each independently pick the in-tree
256x256 MFMA kernel or hipBLASLt from the same-box A/B table below
(gpurun_out/kbench_gemm5/6.log, profiles/r02_notes.md):

  op      mine (TF)      hipBLASLt (TF)   policy
  fprop   1000-1200      1165-1590        library
  dgrad   1075-1145      1250-1390        library
  wgrad   900-1050       835-1190         MINE for the <=8-tile-per-dim
                                          deep-K family (e.g. out-proj
                                          dW 2048x2048xK16384: +8-10%) —
                                          the shape class round 1 measured
                                          hipBLASLt weakest on; library
                                          elsewhere

Env ``TDPA_GEMM``: "1" (default) = the measured auto policy above;
"all" = force every eligible GEMM through the in-tree kernel (kernel
demonstration / profiling); "0" = library everywhere.

The reference delegates all of these to cuBLAS via torch.matmul
(tp_utils.py:171); BASELINE.json's north star names the transformer
projection GEMMs as hand-written CDNA4 — csrc/gemm.hip is that path, and
the auto policy keeps it on the hot step where it is a measured win.
"""

from __future__ import annotations

import os
from typing import Optional

import torch
import torch.nn.functional as F

from . import ext

_MODE = os.environ.get("TDPA_GEMM", "1")
_GEMV = os.environ.get("TDPA_GEMV", "1") == "1"
# wgrad K=16384-class shapes keep fp32 partials in a slab; cap its size
_MAX_SLAB_BYTES = 2 << 30


def _eligible(M: int, N: int, K: int) -> bool:
    return M % 256 == 0 and N % 256 == 0 and K % 32 == 0 and K >= 32


def pick_splitk(M: int, N: int, K: int) -> int:
    """Fill the 256-CU chip: prefer the split factor whose grid is closest
    to a whole multiple of 256 blocks (1 block/CU kernel)."""
    ntiles = (M // 256) * (N // 256)
    best, best_eff = 1, 0.0
    for sk in (1, 2, 4, 8):
        if K % (32 * sk) or K // sk < 128:
            continue
        if sk > 1 and sk * M * N * 4 > _MAX_SLAB_BYTES:
            continue
        blocks = ntiles * sk
        waves = -(-blocks // 256)
        eff = blocks / (256.0 * waves) / (1.0 + 0.03 * (sk - 1))
        if eff > best_eff + 1e-9:
            best, best_eff = sk, eff
    return best


def _use_gemv(M: int, N: int, K: int) -> bool:
    """Measured crossover (scripts/kbench_gemv.py, profiles/r02_gemv.log):
    the streaming GEMV wins while its ~linear-in-M time stays under
    hipBLASLt's flat ~18 us launch floor.  The isolated microbench also
    shows a proj-shape win at M=16, but inside the captured decode graph
    that substitution measured -5% end to end (r02_decode AB) — the
    policy follows the end-to-end number and stops at M=8."""
    if M <= 4 and N <= 16384:
        return True
    return M <= 8 and K <= 4096 and N <= 16384


def _use_mine(kind: str, M: int, N: int, K: int) -> bool:
    if _MODE == "all":
        return True
    if _MODE != "1":
        return False
    if kind == "wgrad":
        # few-tile deep-K wgrad: hipBLASLt's weak family (r01 item 4)
        return M <= 2048 and N <= 2048 and K >= 4096
    return False


def gemm_enabled() -> bool:
    return _MODE != "0"


class _TdpaLinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x2d, weight, bias):
        ctx.save_for_backward(x2d, weight)
        ctx.has_bias = bias is not None
        M, K = x2d.shape
        N = weight.shape[0]
        if _use_mine("fprop", M, N, K):
            return ext("gemm_fprop").gemm_fprop(x2d, weight, bias)
        out = F.linear(x2d, weight, bias)
        return out

    @staticmethod
    def backward(ctx, dy):
        x2d, weight = ctx.saved_tensors
        dy = dy.contiguous()
        M, K = x2d.shape
        N = weight.shape[0]
        # the backward GEMMs additionally need the in-features dim
        # 256-aligned for the in-tree kernel
        if K % 256 == 0 and _use_mine("dgrad", M, K, N):
            dx = ext("gemm").gemm_dgrad(dy, weight, True)
        else:
            dx = dy @ weight
        if K % 256 == 0 and _use_mine("wgrad", N, K, M):
            dw = ext("gemm").gemm_wgrad(dy, x2d, pick_splitk(N, K, M), True)
        else:
            dw = dy.t() @ x2d
        db = dy.sum(0) if ctx.has_bias else None
        return dx, dw, db


def linear(x: torch.Tensor, weight: torch.Tensor,
           bias: Optional[torch.Tensor] = None) -> torch.Tensor:
    """F.linear drop-in; routes 256-aligned bf16 GPU shapes through the
    per-GEMM measured dispatch (in-tree MFMA kernel vs hipBLASLt)."""
    if (_MODE != "0" and x.is_cuda and x.dtype == torch.bfloat16
            and weight.dtype == torch.bfloat16
            and (bias is None or bias.dtype == torch.bfloat16)):
        shape = x.shape
        M = x.numel() // shape[-1]
        K = shape[-1]
        N = weight.shape[0]
        # decode path (small-M tokens, inference only): the in-tree
        # streaming GEMV (csrc/gemv.hip) reads weights at 5-7.6 TB/s where
        # hipBLASLt's skinny-M kernels have a flat ~18 us floor; the kernel's
        # cost grows ~linearly in M (it turns LDS/VALU-bound), so dispatch
        # follows the measured crossover table (profiles/r02_gemv.log).
        # TDPA_GEMV=0 reverts to hipBLASLt everywhere.
        if (_GEMV and K % 8 == 0 and _use_gemv(M, N, K)
                and not torch.is_grad_enabled()
                and weight.is_contiguous()):
            x2d = x.reshape(M, K)
            if not x2d.is_contiguous():
                x2d = x2d.contiguous()
            out = ext("gemv").gemv_bf16(x2d, weight, bias)
            return out.reshape(*shape[:-1], N)
        if _eligible(M, N, K) and (
                _MODE == "all"
                or (K % 256 == 0 and (_use_mine("wgrad", N, K, M)
                                      or _use_mine("dgrad", M, K, N)))
                or _use_mine("fprop", M, N, K)):
            x2d = x.reshape(M, K)
            if not x2d.is_contiguous():
                x2d = x2d.contiguous()
            out = _TdpaLinearFn.apply(x2d, weight.contiguous(), bias)
            return out.reshape(*shape[:-1], N)
    return F.linear(x, weight, bias)
