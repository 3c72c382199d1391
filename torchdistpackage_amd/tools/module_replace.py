"""Recursive module replacement utilities.

Reference parity: /root/reference/torchdistpackage/tools/module_replace.py:1-8
plus the int8 swap-in pattern (bnb_fc.py / bminf_int8.py) generalized: the
int8 libraries the reference wraps (bitsandbytes, bminf) are CUDA-only; the
hooks here are optional-import guarded the same way the reference guards them
(__init__.py:19-24).
"""

from __future__ import annotations

from typing import Callable

import torch.nn as nn


def replace_all_module(model: nn.Module, predicate: Callable[[nn.Module], bool],
                       factory: Callable[[nn.Module], nn.Module]) -> int:
    """Replace every descendant module matching ``predicate`` with
    ``factory(old_module)``.  Returns the number of replacements."""
    count = 0
    for name, child in list(model.named_children()):
        if predicate(child):
            setattr(model, name, factory(child))
            count += 1
        else:
            count += replace_all_module(child, predicate, factory)
    return count


def replace_linear_by_bnb(model: nn.Module, threshold: float = 6.0) -> int:
    """Swap nn.Linear -> bitsandbytes Linear8bitLt (optional dependency)."""
    import bitsandbytes as bnb  # noqa: F401

    def factory(old: nn.Linear):
        new = bnb.nn.Linear8bitLt(old.in_features, old.out_features,
                                  bias=old.bias is not None,
                                  has_fp16_weights=False, threshold=threshold)
        new.weight.data.copy_(old.weight.data)
        if old.bias is not None:
            new.bias.data.copy_(old.bias.data)
        return new

    return replace_all_module(model, lambda m: isinstance(m, nn.Linear), factory)


def replace_linear_by_bminf(model: nn.Module) -> int:
    """Swap nn.Linear -> bminf QuantizedLinear (optional dependency)."""
    import bminf

    def factory(old: nn.Linear):
        return bminf.QuantizedLinear(old)

    return replace_all_module(model, lambda m: isinstance(m, nn.Linear), factory)
