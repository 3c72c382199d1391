"""Distributed init for one-process-per-GPU MI355X training.

Capability parity with the reference ``setup_distributed``
(/root/reference/torchdistpackage/dist/launch_from_slurm.py:16-62): reads either
torchrun env (RANK/WORLD_SIZE/LOCAL_RANK/MASTER_ADDR/MASTER_PORT) or SLURM env
(SLURM_PROCID/SLURM_NTASKS/SLURM_NODELIST), initializes torch.distributed and
pins the HIP device.

Backend "nccl" on ROCm *is* RCCL; over a single 8-GPU MI355X node it runs ring
or direct-p2p collectives over the 7 xGMI links per GPU.  A composite
"cpu:gloo,cuda:nccl" backend is used when CUDA is available so CPU-tensor
barriers don't touch the GPU; plain gloo otherwise (CPU CI).
"""

from __future__ import annotations

import os
import socket
import subprocess
from datetime import timedelta

import torch
import torch.distributed as dist

DEFAULT_PORT = 54647


def find_free_port() -> int:
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("", 0))
        return s.getsockname()[1]


def _slurm_master_addr() -> str:
    nodelist = os.environ["SLURM_NODELIST"]
    try:
        out = subprocess.check_output(
            ["scontrol", "show", "hostname", nodelist], text=True)
        return out.split()[0]
    except (OSError, subprocess.CalledProcessError):
        # crude fallback: "host[1-4]" -> "host1"
        return nodelist.split(",")[0].replace("[", "").split("-")[0]


def setup_distributed(backend: str = None, port: int = None,
                      timeout_s: int = 1800) -> dict:
    """Initialize torch.distributed from torchrun or SLURM env.

    Returns a dict with rank / world_size / local_rank / master_addr.
    Safe to call when already initialized (no-op then).
    """
    if dist.is_initialized():
        return {
            "rank": dist.get_rank(),
            "world_size": dist.get_world_size(),
            "local_rank": int(os.environ.get("LOCAL_RANK", 0)),
            "master_addr": os.environ.get("MASTER_ADDR", "127.0.0.1"),
        }

    if "RANK" in os.environ and "WORLD_SIZE" in os.environ:
        rank = int(os.environ["RANK"])
        world_size = int(os.environ["WORLD_SIZE"])
        local_rank = int(os.environ.get("LOCAL_RANK", rank))
        master_addr = os.environ.get("MASTER_ADDR", "127.0.0.1")
        master_port = int(os.environ.get("MASTER_PORT", port or DEFAULT_PORT))
    elif "SLURM_PROCID" in os.environ:
        rank = int(os.environ["SLURM_PROCID"])
        world_size = int(os.environ["SLURM_NTASKS"])
        local_rank = rank % max(torch.cuda.device_count(), 1)
        master_addr = _slurm_master_addr()
        master_port = port or DEFAULT_PORT
        os.environ["MASTER_ADDR"] = master_addr
        os.environ["MASTER_PORT"] = str(master_port)
        os.environ["RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world_size)
        os.environ["LOCAL_RANK"] = str(local_rank)
    else:
        # single-process fallback
        rank, world_size, local_rank = 0, 1, 0
        master_addr = "127.0.0.1"
        master_port = port or find_free_port()
        os.environ.setdefault("MASTER_ADDR", master_addr)
        os.environ.setdefault("MASTER_PORT", str(master_port))
        os.environ.setdefault("RANK", "0")
        os.environ.setdefault("WORLD_SIZE", "1")

    if backend is None:
        backend = "cpu:gloo,cuda:nccl" if torch.cuda.is_available() else "gloo"

    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank % torch.cuda.device_count())

    dist.init_process_group(
        backend=backend,
        rank=rank,
        world_size=world_size,
        timeout=timedelta(seconds=timeout_s),
    )
    return {
        "rank": rank,
        "world_size": world_size,
        "local_rank": local_rank,
        "master_addr": master_addr,
    }
