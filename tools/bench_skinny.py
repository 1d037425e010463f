#!/usr/bin/env python3
"""A/B the weight-streaming skinny GEMM vs torch.matmul (hipBLASLt) on the
7B decode shapes.  Run on an MI355X."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch

import realhf_amd._C as C

SHAPES = [  # (name, N, K)
    ("qkv", 12288, 4096),
    ("o", 4096, 4096),
    ("gateup", 22016, 4096),
    ("down", 4096, 11008),
    ("lmhead", 32000, 4096),
]


def pick_splitk(N, K):
    sk = 1
    while (N // 64) * sk < 512 and sk < 16:
        sk *= 2
    while K // sk > 1024:
        sk *= 2
    return sk


def bench(fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    M = 16
    ws = torch.empty(32 * M * 32000, dtype=torch.float32, device="cuda")
    sem = torch.zeros(1024, dtype=torch.int32, device="cuda")
    print(f"{'shape':8} {'N':>6} {'K':>6} {'sk':>3} {'blaslt_us':>9} "
          f"{'skinny_us':>9} {'v2_us':>7} {'roofline_us':>11} "
          f"{'max_err':>8} {'v2_err':>8}")
    for name, N, K in SHAPES:
        x = (torch.randn(M, K, device="cuda") * 0.3).to(torch.bfloat16)
        w = (torch.randn(N, K, device="cuda") * 0.3).to(torch.bfloat16)
        ref = x.float() @ w.float().t()
        t_blas = bench(lambda: torch.matmul(x, w.t()))
        roof = N * K * 2 / 6.3e12 * 1e6
        for sk in sorted({pick_splitk(N, K), 2, 4, 8, 16}):
            if K // sk > 1024 or K % 32:
                continue
            out = C.skinny_gemm(x, w, ws, sk, None)
            err = (out.float() - ref).abs().max().item() / ref.abs().max().item()
            out2 = C.skinny_gemm2(x, w, ws, sem, sk, None)
            err2 = (out2.float() - ref).abs().max().item() / ref.abs().max().item()
            t_sk = bench(lambda: C.skinny_gemm(x, w, ws, sk, None))
            t_v2 = bench(lambda: C.skinny_gemm2(x, w, ws, sem, sk, None))
            print(f"{name:8} {N:6d} {K:6d} {sk:3d} {t_blas:9.1f} {t_sk:9.1f} "
                  f"{t_v2:7.1f} {roof:11.1f} {err:8.4f} {err2:8.4f}")


if __name__ == "__main__":
    main()
