"""CPU unit tests for the GEMM dispatch policy (ops/gemm.py) — the pure
host-side logic: shape eligibility, split-K chip-fill selection, and the
TDPA_GEMM env modes.  The kernels themselves are covered in test_ops_gpu."""
import importlib
import os

import pytest
import torch

import torchdistpackage_amd.ops.gemm as gemm_mod


def _reload(mode):
    if mode is None:
        os.environ.pop("TDPA_GEMM", None)
    else:
        os.environ["TDPA_GEMM"] = mode
    return importlib.reload(gemm_mod)


@pytest.fixture(autouse=True)
def _restore_mode():
    prev = os.environ.get("TDPA_GEMM")
    yield
    if prev is None:
        os.environ.pop("TDPA_GEMM", None)
    else:
        os.environ["TDPA_GEMM"] = prev
    importlib.reload(gemm_mod)


def test_eligibility():
    g = _reload(None)
    assert g._eligible(256, 256, 32)
    assert g._eligible(2048, 2048, 16384)
    assert not g._eligible(255, 256, 32)    # M misaligned
    assert not g._eligible(256, 192, 32)    # N misaligned
    assert not g._eligible(256, 256, 48)    # K not /32


def test_pick_splitk_invariants():
    g = _reload(None)
    for M, N, K in [(2048, 2048, 16384), (8192, 8192, 4096),
                    (256, 256, 128), (4096, 4096, 8192),
                    (512, 2048, 4096)]:
        sk = g.pick_splitk(M, N, K)
        assert K % (32 * sk) == 0
        assert K // sk >= 128
        assert sk == 1 or sk * M * N * 4 <= g._MAX_SLAB_BYTES


def test_pick_splitk_fills_chip():
    g = _reload(None)
    # out-proj dW (2048x2048, 64 tiles): sk=4 lands exactly on 256 blocks
    assert g.pick_splitk(2048, 2048, 16384) == 4
    # 1024 tiles already a whole multiple of 256 blocks: no split
    assert g.pick_splitk(8192, 8192, 4096) == 1


def test_env_modes():
    g = _reload("0")
    assert not g.gemm_enabled()
    assert not g._use_mine("wgrad", 2048, 2048, 16384)

    g = _reload("1")
    assert g.gemm_enabled()
    # the measured hipBLASLt-weak family: few tiles, deep K
    assert g._use_mine("wgrad", 2048, 2048, 16384)
    assert g._use_mine("wgrad", 768, 768, 4096)
    # library keeps shallow-K / many-tile wgrad and all fprop/dgrad
    assert not g._use_mine("wgrad", 2048, 2048, 2048)
    assert not g._use_mine("wgrad", 4096, 2048, 16384)
    assert not g._use_mine("fprop", 2048, 2048, 16384)
    assert not g._use_mine("dgrad", 2048, 2048, 16384)

    g = _reload("all")
    for kind in ("fprop", "dgrad", "wgrad"):
        assert g._use_mine(kind, 256, 256, 32)


def test_linear_cpu_fallback_exact():
    g = _reload("1")
    torch.manual_seed(0)
    x = torch.randn(4, 256, 512)
    w = torch.randn(768, 512)
    b = torch.randn(768)
    assert torch.equal(g.linear(x, w, b),
                       torch.nn.functional.linear(x, w, b))


def test_gemv_crossover_policy():
    g = _reload("1")
    # batch-1 latency decode: every non-head projection routes to the GEMV
    assert g._use_gemv(1, 6144, 2048)
    assert g._use_gemv(1, 2048, 8192)
    assert g._use_gemv(4, 8192, 2048)
    assert g._use_gemv(8, 6144, 2048)     # K<=4096 family holds to M=8
    assert not g._use_gemv(8, 2048, 8192)  # deep-K falls off after M=4
    assert not g._use_gemv(16, 2048, 2048)  # in-graph AB: library wins
    assert not g._use_gemv(1, 50304, 2048)  # vocab head: library
