"""Reproducibility + small shared helpers.

Reference parity: /root/reference/torchdistpackage/utils.py:4-33 (fix_rand)
and :35-65 (greedy numel partition, re-exported from dist.sharded_ema).
"""

from __future__ import annotations

import os
import random

import numpy as np
import torch

from .dist.sharded_ema import partition_by_numel  # noqa: F401  (re-export)


def fix_rand(rank: int = 0, seed: int = 2222, deterministic: bool = True):
    """Seed torch / HIP / numpy / python with seed+rank; optionally force
    deterministic algorithms.

    On ROCm the cudnn flags map to MIOpen's find-mode determinism; unlike the
    reference (utils.py:24 disables cudnn entirely) we keep MIOpen enabled —
    determinism comes from ``benchmark=False`` + ``deterministic=True``.
    """
    s = seed + rank
    random.seed(s)
    np.random.seed(s)
    torch.manual_seed(s)
    os.environ["PYTHONHASHSEED"] = str(s)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(s)
    if deterministic:
        torch.backends.cudnn.benchmark = False
        torch.backends.cudnn.deterministic = True
