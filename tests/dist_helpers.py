"""Spawn-based multi-process test harness: run a function under world_size N
ranks with gloo (CPU) or nccl/RCCL (GPU).

Used by every distributed CPU test (BASELINE.json config 1: "NaiveDdp on
2-layer MLP, gloo world_size=2 on CPU").
"""

from __future__ import annotations

import os
import pickle
import tempfile
import traceback
from datetime import timedelta

import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def _worker(rank, world_size, port, fn, args, kwargs, backend, result_dir):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ["RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world_size)
        os.environ["LOCAL_RANK"] = str(rank)
        if backend == "nccl":
            torch.cuda.set_device(rank % torch.cuda.device_count())
        dist.init_process_group(backend=backend, rank=rank,
                                world_size=world_size,
                                timeout=timedelta(seconds=120))
        # fresh topology singleton per process (spawn gives us that for free)
        out = fn(rank, world_size, *args, **kwargs)
        with open(os.path.join(result_dir, f"rank{rank}.pkl"), "wb") as f:
            pickle.dump(("ok", out), f)
    except Exception:
        with open(os.path.join(result_dir, f"rank{rank}.pkl"), "wb") as f:
            pickle.dump(("err", traceback.format_exc()), f)
        raise
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def run_distributed(fn, world_size: int = 2, backend: str = "gloo",
                    args=(), kwargs=None, timeout: float = 180.0):
    """Run ``fn(rank, world_size, *args, **kwargs)`` in ``world_size``
    spawned processes; returns the list of per-rank return values."""
    kwargs = kwargs or {}
    from torchdistpackage_amd.dist.launch import find_free_port
    last_exc = None
    for attempt in range(2):   # one retry for infra flakes (port races,
        port = find_free_port()  # spawn hiccups under parallel CI load)
        with tempfile.TemporaryDirectory() as result_dir:
            try:
                mp.spawn(
                    _worker,
                    args=(world_size, port, fn, args, kwargs, backend,
                          result_dir),
                    nprocs=world_size, join=True, daemon=False)
            except Exception as e:  # noqa: BLE001
                # a REAL test failure leaves a pickled traceback; surface it
                for r in range(world_size):
                    path = os.path.join(result_dir, f"rank{r}.pkl")
                    if os.path.exists(path):
                        with open(path, "rb") as f:
                            status, payload = pickle.load(f)
                        if status == "err":
                            raise AssertionError(
                                f"rank {r} failed:\n{payload}") from e
                last_exc = e
                continue  # infra-level failure: retry once on a new port
            results = []
            for r in range(world_size):
                path = os.path.join(result_dir, f"rank{r}.pkl")
                assert os.path.exists(path), f"rank {r} produced no result"
                with open(path, "rb") as f:
                    status, payload = pickle.load(f)
                assert status == "ok", f"rank {r} failed:\n{payload}"
                results.append(payload)
            return results
    raise last_exc
