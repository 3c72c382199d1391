"""Model-parallel checkpoint filename helper.

Reference parity: /root/reference/torchdistpackage/dist/model_parallel_ckpt.py
(which is buggy as written — calls an unqualified ``is_mode_inited``; fixed
here by querying ``tpc`` properly).
"""

from __future__ import annotations

from .topo import tpc


def get_mp_ckpt_suffix() -> str:
    """Suffix ``_tp_{r}_pp_{r}`` for whatever model-parallel axes are active."""
    parts = []
    if tpc.is_mode_inited("tensor") and tpc.get_tp_size() > 1:
        parts.append(f"_tp_{tpc.get_tp_rank()}")
    if tpc.is_mode_inited("pipe") and tpc.get_pp_size() > 1:
        parts.append(f"_pp_{tpc.get_pp_rank()}")
    return "".join(parts)


def mp_ckpt_name(base: str, ext: str = ".pth") -> str:
    return f"{base}{get_mp_ckpt_suffix()}{ext}"
