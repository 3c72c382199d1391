"""FusedAdamW: AdamW on the in-tree HIP kernel, with fp32 master weights for
bf16 params.

Replaces the reference's reliance on stock torch.optim.Adam/AdamW
(/root/reference/torchdistpackage/ddp/zero_optim.py:265, examples).  Per-param
state lives in fp32; on GPU the update is one fused kernel per param tensor
(param/exp_avg/exp_avg_sq read+write in a single HBM pass).
"""

from __future__ import annotations

from typing import Optional

import torch

from . import fused_adamw_


class FusedAdamW(torch.optim.Optimizer):
    def __init__(self, params, lr: float = 1e-3, betas=(0.9, 0.95),
                 eps: float = 1e-8, weight_decay: float = 0.01):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            lr = group["lr"]
            beta1, beta2 = group["betas"]
            eps = group["eps"]
            wd = group["weight_decay"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["exp_avg"] = torch.zeros_like(
                        p, dtype=torch.float32)
                    state["exp_avg_sq"] = torch.zeros_like(
                        p, dtype=torch.float32)
                    if p.dtype != torch.float32:
                        state["master"] = p.detach().float().clone()
                state["step"] += 1
                grad = p.grad
                if grad.dtype != torch.float32:
                    grad = grad.float()
                target = state.get("master", p)
                fused_adamw_(target.view(-1), grad.contiguous().view(-1),
                             state["exp_avg"].view(-1),
                             state["exp_avg_sq"].view(-1),
                             state["step"], lr, beta1, beta2, eps, wd)
                if "master" in state:
                    p.copy_(target.to(p.dtype))
        return loss
