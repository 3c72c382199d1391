"""GEMM kernel micro-benchmark + numerics check vs hipBLASLt (torch.matmul).

Run on the GPU box:
    python scripts/kbench_gemm.py            # correctness + bench shapes
    python scripts/kbench_gemm.py --quick    # correctness only

Shapes = the GPT-2 1.3B bench-step hot GEMMs (B=16, S=1024 -> 16384 tokens)
incl. the wgrad K=16384 family hipBLASLt runs at ~1.03 PF/s
(profiles/r01_notes.md item 4).
"""

import argparse
import sys

import torch

sys.path.insert(0, ".")
from torchdistpackage_amd.ops import ext  # noqa: E402
from torchdistpackage_amd.ops.gemm import pick_splitk  # noqa: E402


def relerr(out, ref):
    return ((out.float() - ref).abs().max() / ref.abs().max()).item()


def check(name, out, ref, yard):
    e = relerr(out, ref)
    ok = e <= max(2.5 * yard, 1e-3)
    print(f"  {name:28s} relerr={e:.2e} (hipblaslt yardstick {yard:.2e}) "
          f"{'OK' if ok else 'FAIL'}")
    return ok


def correctness():
    e = ext("gemm")
    torch.manual_seed(0)
    ok = True
    for (M, N, K) in [(512, 512, 512), (256, 256, 64), (512, 256, 160),
                      (768, 512, 1024)]:
        x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16) * 0.5
        w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16) * 0.5
        b = torch.randn(N, device="cuda", dtype=torch.bfloat16)
        ref = x.float() @ w.float().t()
        yard = relerr(x @ w.t(), ref)
        ok &= check(f"fprop {M}x{N}x{K}", e.gemm_fprop(x, w, None), ref, yard)
        ok &= check(f"fprop+bias {M}x{N}x{K}",
                    e.gemm_fprop(x, w, b), ref + b.float(), yard + 1e-3)
        # dgrad: dy (M,N) @ w2 (N,K2)
        if N % 32 == 0 and K % 256 == 0:
            w2 = torch.randn(N, K, device="cuda", dtype=torch.bfloat16) * 0.5
            dy = torch.randn(M, N, device="cuda", dtype=torch.bfloat16) * 0.5
            ref2 = dy.float() @ w2.float()
            yard2 = relerr(dy @ w2, ref2)
            for sw in (False, True):
                ok &= check(f"dgrad{'S' if sw else ' '} {M}x{K}x{N}",
                            e.gemm_dgrad(dy, w2, sw), ref2, yard2)
        # wgrad: dy (T,M) ^T @ x (T,N); T = K here
        if K % 32 == 0:
            T = max(K, 512)
            dyt = torch.randn(T, M, device="cuda", dtype=torch.bfloat16) * 0.5
            xt = torch.randn(T, N, device="cuda", dtype=torch.bfloat16) * 0.5
            ref3 = dyt.float().t() @ xt.float()
            yard3 = relerr(dyt.t() @ xt, ref3)
            for sk in (1, 2):
                if T % (32 * sk):
                    continue
                for sw in (False, True):
                    ok &= check(f"wgrad{'S' if sw else ' '}/sk{sk} "
                                f"{M}x{N}xT{T}",
                                e.gemm_wgrad(dyt, xt, sk, sw), ref3, yard3)
    print("correctness:", "ALL OK" if ok else "FAILURES")
    return ok


def bench_one(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    t = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    t.record()
    torch.cuda.synchronize()
    return s.elapsed_time(t) / iters * 1e-3  # seconds


def bench():
    e = ext("gemm")
    torch.manual_seed(1)
    T = 16384  # tokens at the bench shape
    rows = []
    # (tag, kind, M, N, K)
    shapes = [
        ("qkv-fprop", "fprop", T, 6144, 2048),
        ("proj-fprop", "fprop", T, 2048, 2048),
        ("fc1-fprop", "fprop", T, 8192, 2048),
        ("fc2-fprop", "fprop", T, 2048, 8192),
        ("qkv-dgrad", "dgrad", T, 2048, 6144),
        ("fc2-dgrad", "dgrad", T, 8192, 2048),
        ("qkv-wgrad", "wgrad", 6144, 2048, T),
        ("proj-wgrad", "wgrad", 2048, 2048, T),
        ("fc1-wgrad", "wgrad", 8192, 2048, T),
        ("fc2-wgrad", "wgrad", 2048, 8192, T),
    ]
    for tag, kind, M, N, K in shapes:
        flops = 2.0 * M * N * K
        if kind == "fprop":
            x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
            w = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
            mine = lambda: e.gemm_fprop(x, w, None)
            lib = lambda: x @ w.t()
        elif kind == "dgrad":
            dy = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
            w = torch.randn(K, N, device="cuda", dtype=torch.bfloat16)
            mine = lambda: e.gemm_dgrad(dy, w, False)
            mine_sw = lambda: e.gemm_dgrad(dy, w, True)
            lib = lambda: dy @ w
        else:
            dy = torch.randn(K, M, device="cuda", dtype=torch.bfloat16)
            x = torch.randn(K, N, device="cuda", dtype=torch.bfloat16)
            sk = pick_splitk(M, N, K)
            mine = lambda: e.gemm_wgrad(dy, x, sk, False)
            mine_sw = lambda: e.gemm_wgrad(dy, x, sk, True)
            lib = lambda: dy.t() @ x
        t_lib = bench_one(lib)
        t_mine = bench_one(mine)
        extra = ""
        if kind != "fprop":
            t_sw = bench_one(mine_sw)
            extra = f" kswz={flops/t_sw/1e12:7.0f}TF"
        sk_s = f" sk={pick_splitk(M, N, K)}" if kind == "wgrad" else ""
        print(f"{tag:12s} {M}x{N}x{K}{sk_s}: "
              f"mine={flops/t_mine/1e12:7.0f}TF "
              f"hipblaslt={flops/t_lib/1e12:7.0f}TF "
              f"ratio={t_lib/t_mine:5.2f}x{extra}")
        rows.append((tag, flops / t_mine / 1e12, flops / t_lib / 1e12))
    return rows


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--quick", action="store_true")
    args = ap.parse_args()
    ok = correctness()
    if not args.quick:
        bench()
    sys.exit(0 if ok else 1)
