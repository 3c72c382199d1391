"""Checkpoint/resume orchestration for mixed-parallel training.

The reference has only fragments here (a buggy MP filename
helper, EMA shard state dicts, scaler state) — no model/optimizer save-load
orchestration.  This module provides it:

- ``save_checkpoint`` / ``load_checkpoint``: model + optimizer (+EMA, +scaler,
  +RNG states, +user extras) into a directory, one file per MP shard
  (``_tp_{r}_pp_{r}`` suffix via mp_ckpt) written by dp-rank-0 of each model
  shard only.
- DP-replicated tensors are written once; TP/PP-sharded state is written per
  shard rank and loaded back by the same topology.
"""

from __future__ import annotations

import os
import random
from typing import Any, Dict, Optional

import numpy as np
import torch
import torch.distributed as dist

from .mp_ckpt import get_mp_ckpt_suffix
from .topo import tpc


def _rng_state() -> Dict[str, Any]:
    st = {
        "torch": torch.get_rng_state(),
        "numpy": np.random.get_state(),
        "python": random.getstate(),
    }
    if torch.cuda.is_available():
        st["hip"] = torch.cuda.get_rng_state()
    return st


def _load_rng_state(st: Dict[str, Any]):
    torch.set_rng_state(st["torch"])
    np.random.set_state(st["numpy"])
    random.setstate(st["python"])
    if torch.cuda.is_available() and "hip" in st:
        torch.cuda.set_rng_state(st["hip"])


def _is_save_rank() -> bool:
    """dp-rank-0 of this model shard writes; everyone else skips."""
    if not dist.is_initialized():
        return True
    if tpc.is_mode_inited("data"):
        return tpc.get_dp_rank() == 0
    return dist.get_rank() == 0


def _dp_rank() -> int:
    if not dist.is_initialized():
        return 0
    if tpc.is_mode_inited("data"):
        return tpc.get_dp_rank()
    return dist.get_rank()


def save_checkpoint(directory: str, step: int, model: torch.nn.Module,
                    optimizer=None, ema=None, scaler=None,
                    extra: Optional[Dict[str, Any]] = None,
                    save_rng: bool = True):
    """Write ``{directory}/ckpt_step{step}{mp_suffix}.pth`` from each model
    shard's dp-rank-0; rank 0 also writes a ``latest`` pointer file.

    Optimizers that mark themselves ``sharded_state = True`` (ZeRO: each dp
    rank owns a distinct master/optimizer shard) are written as one
    ``optim_step{N}{mp_suffix}_dp{r}.pth`` file PER dp rank instead of being
    embedded in dp-rank-0's payload."""
    os.makedirs(directory, exist_ok=True)
    sharded_opt = optimizer is not None and \
        getattr(optimizer, "sharded_state", False)
    if sharded_opt:
        oname = f"optim_step{step}{get_mp_ckpt_suffix()}_dp{_dp_rank()}.pth"
        torch.save({"optimizer": optimizer.state_dict()},
                   os.path.join(directory, oname))
    if _is_save_rank():
        payload: Dict[str, Any] = {
            "step": step,
            "model": model.state_dict(),
        }
        if optimizer is not None and not sharded_opt:
            payload["optimizer"] = optimizer.state_dict()
        if scaler is not None:
            payload["scaler"] = scaler.state_dict()
        if save_rng:
            payload["rng"] = _rng_state()
        if extra:
            payload["extra"] = extra
        name = f"ckpt_step{step}{get_mp_ckpt_suffix()}.pth"
        torch.save(payload, os.path.join(directory, name))
    if ema is not None:
        # EMA is dp-sharded: reassemble on group rank 0 and save there
        full = ema.state_dict_cpu()
        if full is not None and (not dist.is_initialized() or
                                 _is_save_rank()):
            torch.save(full, os.path.join(
                directory, f"ema_step{step}{get_mp_ckpt_suffix()}.pth"))
    # all shard files must exist before 'latest' moves: barrier first, then
    # rank 0 publishes via fsync'd tmp file + atomic rename — a crash mid-save
    # can never leave 'latest' pointing at an incomplete checkpoint set
    if dist.is_initialized():
        dist.barrier()
    if not dist.is_initialized() or dist.get_rank() == 0:
        tmp = os.path.join(directory, "latest.tmp")
        with open(tmp, "w") as f:
            f.write(str(step))
            f.flush()
            os.fsync(f.fileno())
        os.replace(tmp, os.path.join(directory, "latest"))
    if dist.is_initialized():
        dist.barrier()


def latest_step(directory: str) -> Optional[int]:
    p = os.path.join(directory, "latest")
    if not os.path.exists(p):
        return None
    with open(p) as f:
        return int(f.read().strip())


def load_checkpoint(directory: str, model: torch.nn.Module,
                    optimizer=None, scaler=None, ema=None,
                    step: Optional[int] = None,
                    map_location="cpu", load_rng: bool = True,
                    strict: bool = True) -> Dict[str, Any]:
    """Load this rank's MP shard of the checkpoint; returns the payload."""
    if step is None:
        step = latest_step(directory)
        if step is None:
            raise FileNotFoundError(f"no 'latest' pointer in {directory}")
    name = f"ckpt_step{step}{get_mp_ckpt_suffix()}.pth"
    payload = torch.load(os.path.join(directory, name),
                         map_location=map_location, weights_only=False)
    model.load_state_dict(payload["model"], strict=strict)
    if optimizer is not None and getattr(optimizer, "sharded_state", False):
        oname = f"optim_step{step}{get_mp_ckpt_suffix()}_dp{_dp_rank()}.pth"
        osd = torch.load(os.path.join(directory, oname),
                         map_location=map_location, weights_only=False)
        optimizer.load_state_dict(osd["optimizer"])
    elif optimizer is not None and "optimizer" in payload:
        optimizer.load_state_dict(payload["optimizer"])
    if scaler is not None and "scaler" in payload:
        scaler.load_state_dict(payload["scaler"])
    if ema is not None:
        ename = f"ema_step{step}{get_mp_ckpt_suffix()}.pth"
        full = torch.load(os.path.join(directory, ename),
                          map_location=map_location, weights_only=False)
        ema.load_state_dict(full)
    if load_rng and "rng" in payload:
        _load_rng_state(payload["rng"])
    if dist.is_initialized():
        dist.barrier()
    return payload
