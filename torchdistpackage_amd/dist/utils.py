"""Profiling-window + misc distributed utilities (MI355X / ROCm).

Reference parity: /root/reference/torchdistpackage/dist/utils.py (NVTX ranges,
cudaProfilerStart/Stop windows, nan/inf scan, master-only print).

On ROCm, ``torch.cuda.nvtx`` maps to rocTX and ``torch.cuda.cudart``'s profiler
start/stop map to the roctracer window API, so rocprofv3's ``--trace-period``
/ marker filtering can key off the same calls.
"""

from __future__ import annotations

import builtins
import functools
import time

import torch
import torch.distributed as dist


def hip_prof_start():
    """Open a profiler capture window (rocprof/roctracer)."""
    if torch.cuda.is_available():
        torch.cuda.synchronize()
        torch.cuda.cudart().cudaProfilerStart()


def hip_prof_stop():
    if torch.cuda.is_available():
        torch.cuda.synchronize()
        torch.cuda.cudart().cudaProfilerStop()


# reference-compatible aliases
cu_prof_start = hip_prof_start
cu_prof_stop = hip_prof_stop


def roctx_decorator(name: str = None, timing: bool = False):
    """Wrap a fn in a rocTX range; optionally synchronize + time it."""

    def deco(fn):
        label = name or fn.__qualname__

        @functools.wraps(fn)
        def wrapper(*args, **kwargs):
            if torch.cuda.is_available():
                torch.cuda.nvtx.range_push(label)
            t0 = time.perf_counter() if timing else None
            try:
                return fn(*args, **kwargs)
            finally:
                if torch.cuda.is_available():
                    torch.cuda.nvtx.range_pop()
                if timing:
                    if torch.cuda.is_available():
                        torch.cuda.synchronize()
                    print(f"[roctx] {label}: "
                          f"{(time.perf_counter() - t0) * 1e3:.3f} ms")
        return wrapper
    return deco


nvtx_decorator = roctx_decorator


class ROCTXContext:
    """``with ROCTXContext("fwd"):`` pushes a rocTX range around the block."""

    def __init__(self, name: str, timing: bool = False):
        self.name = name
        self.timing = timing
        self._t0 = None

    def __enter__(self):
        if torch.cuda.is_available():
            torch.cuda.nvtx.range_push(self.name)
        if self.timing:
            self._t0 = time.perf_counter()
        return self

    def __exit__(self, *exc):
        if torch.cuda.is_available():
            torch.cuda.nvtx.range_pop()
        if self.timing:
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            print(f"[roctx] {self.name}: "
                  f"{(time.perf_counter() - self._t0) * 1e3:.3f} ms")


NVTXContext = ROCTXContext


def has_inf_or_nan(t: torch.Tensor) -> bool:
    """Cheap single-sync scan; sum() is inf/nan iff any element is."""
    s = t.float().sum()
    return bool(torch.isinf(s) | torch.isnan(s))


_orig_print = builtins.print


def disable_non_master_print(force: bool = False):
    """Patch builtins.print so only global rank 0 prints (unless force=True
    is passed to a print call)."""

    def patched(*args, **kwargs):
        f = kwargs.pop("force", False)
        if f or not dist.is_initialized() or dist.get_rank() == 0:
            _orig_print(*args, **kwargs)

    builtins.print = patched


def restore_print():
    builtins.print = _orig_print
