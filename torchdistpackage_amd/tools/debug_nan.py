"""NaN/Inf debugging hooks.

Reference parity: /root/reference/torchdistpackage/tools/debug_nan.py:3-52 —
fwd/bwd hooks scanning outputs (tensors, tuples, ``.sample``-bearing objects)
for nan/inf.  Instead of dropping into pdb (useless under a distributed
launcher) the default action raises with the offending module's name; pass
``action='pdb'`` for the reference behaviour or ``action='print'``.
"""

from __future__ import annotations

from typing import List

import torch
import torch.nn as nn


def _scan(obj) -> bool:
    if torch.is_tensor(obj):
        if obj.is_floating_point():
            s = obj.float().sum()
            return bool(torch.isnan(s) | torch.isinf(s))
        return False
    if isinstance(obj, (list, tuple)):
        return any(_scan(o) for o in obj)
    if hasattr(obj, "sample"):
        return _scan(obj.sample)
    return False


def _act(where: str, name: str, action: str):
    msg = f"[debug_nan] nan/inf detected in {where} of module '{name}'"
    if action == "raise":
        raise FloatingPointError(msg)
    if action == "pdb":
        print(msg)
        import pdb
        pdb.set_trace()
    else:
        print(msg)


def register_nan_hooks(model: nn.Module, action: str = "raise",
                       backward: bool = True) -> List:
    handles = []
    for name, mod in model.named_modules():
        def fwd_hook(m, inp, out, _n=name):
            if _scan(out):
                _act("forward output", _n, action)

        handles.append(mod.register_forward_hook(fwd_hook))
        if backward:
            def bwd_hook(m, gin, gout, _n=name):
                if _scan(gout):
                    _act("backward grad", _n, action)

            handles.append(mod.register_full_backward_hook(bwd_hook))
    return handles


def check_model_params(model: nn.Module, action: str = "raise"):
    """Scan all params (and grads) for nan/inf."""
    for name, p in model.named_parameters():
        if _scan(p):
            _act("param", name, action)
        if p.grad is not None and _scan(p.grad):
            _act("param grad", name, action)
