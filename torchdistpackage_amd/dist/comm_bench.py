"""Collective bus-bandwidth micro-benchmark (nccl-tests busbw convention).

Reference parity: /root/reference/torchdistpackage/dist/py_comm_test.py.
busbw = algbw * frac * (n-1)/n with frac: all_reduce=2, all_gather=1,
reduce_scatter=1, all_to_all=1 (algbw = payload bytes / time).

On one 8×MI355X node the per-GPU xGMI fabric is 7 p2p links × ≈153 GB/s; a
ring collective is bound by one link, so busbw ceilings ≈150 GB/s unless RCCL
uses multi-ring/direct algorithms.  Use this benchmark to pick DDP/ZeRO bucket
sizes (the knee of the busbw-vs-size curve).
"""

from __future__ import annotations

import time
from typing import Dict, List, Optional

import torch
import torch.distributed as dist


def _timeit(fn, iters: int, warmup: int) -> float:
    for _ in range(warmup):
        fn()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def bench_collectives(numel: int = 2 ** 26, dtype=torch.bfloat16,
                      group: Optional[dist.ProcessGroup] = None,
                      iters: int = 20, warmup: int = 5,
                      collectives: Optional[List[str]] = None) -> Dict[str, dict]:
    """Measure alg/bus bandwidth of the standard collectives.

    Returns {name: {time_s, algbw_GBps, busbw_GBps, bytes}} on every rank.
    """
    n = dist.get_world_size(group)
    dev = torch.device("cuda", torch.cuda.current_device()) \
        if torch.cuda.is_available() else torch.device("cpu")
    if dev.type == "cpu" and dtype == torch.bfloat16:
        dtype = torch.float32
    esize = torch.tensor([], dtype=dtype).element_size()
    numel = (numel // (n * 64)) * (n * 64)  # divisible for RS/AG/A2A
    results: Dict[str, dict] = {}
    names = collectives or ["all_reduce", "all_gather", "reduce_scatter",
                            "all_to_all"]

    full = torch.randn(numel, device=dev).to(dtype)
    shard = torch.randn(numel // n, device=dev).to(dtype)
    gathered = torch.empty(numel, dtype=dtype, device=dev)
    a2a_out = torch.empty(numel, dtype=dtype, device=dev)

    frac = {"all_reduce": 2.0, "all_gather": 1.0, "reduce_scatter": 1.0,
            "all_to_all": 1.0}
    ops = {
        "all_reduce": lambda: dist.all_reduce(full, group=group),
        "all_gather": lambda: dist.all_gather_into_tensor(
            gathered, shard, group=group),
        "reduce_scatter": lambda: dist.reduce_scatter_tensor(
            shard, full, group=group),
        "all_to_all": lambda: dist.all_to_all_single(
            a2a_out, full, group=group),
    }

    for name in names:
        if dev.type == "cpu" and name in ("reduce_scatter", "all_to_all") \
                and dist.get_backend(group) == "gloo":
            continue  # gloo lacks these
        t = _timeit(ops[name], iters, warmup)
        payload = numel * esize if name != "all_gather" else numel * esize
        algbw = payload / t / 1e9
        busbw = algbw * frac[name] * (n - 1) / n
        results[name] = {"time_s": t, "algbw_GBps": algbw,
                         "busbw_GBps": busbw, "bytes": payload}
    return results


def main():
    from .launch import setup_distributed
    setup_distributed()
    res = bench_collectives()
    if dist.get_rank() == 0:
        for k, v in res.items():
            print(f"{k:>16}: {v['time_s']*1e3:8.3f} ms  "
                  f"algbw {v['algbw_GBps']:7.1f} GB/s  "
                  f"busbw {v['busbw_GBps']:7.1f} GB/s")


if __name__ == "__main__":
    main()
