#!/usr/bin/env python3
"""Standalone A/B + profiling target for the flash-attention varlen
forward kernel."""
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import realhf_amd._C as C


def bench(fn, iters=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def run(shape_name, lens, nq, nkv, hd=128, iters=30):
    total = sum(lens)
    cu = torch.tensor([0] + list(np.cumsum(lens)), dtype=torch.int32,
                      device="cuda")
    torch.manual_seed(0)
    q = (torch.randn(total, nq, hd, device="cuda") * 0.5).to(torch.bfloat16)
    k = (torch.randn(total, nkv, hd, device="cuda") * 0.5).to(torch.bfloat16)
    v = (torch.randn(total, nkv, hd, device="cuda") * 0.5).to(torch.bfloat16)
    scale = hd ** -0.5
    us = bench(lambda: C.attn_varlen_fwd(q, k, v, cu, max(lens), True, scale, 0),
               iters)
    flops = sum(2 * 2 * (l * l / 2) * nq * hd for l in lens)
    print(f"{shape_name:24} {us:9.1f} us  {flops / us / 1e6:7.1f} TF/s")


if __name__ == "__main__":
    iters = int(sys.argv[1]) if len(sys.argv) > 1 else 30
    run("bench-mb 4x640 h32", [640] * 4, 32, 32, iters=iters)
    run("bench-full 16x640 h32", [640] * 16, 32, 32, iters=iters)
    run("long 16x2048 h32gqa8", [2048] * 16, 32, 8, iters=iters)
    # long-context tier (device-built block lists; grad-ckpt training
    # shapes for the CP/packing long-context story)
    run("16k 2x16384 h32gqa8", [16384] * 2, 32, 8, iters=max(3, iters // 6))
    run("32k 1x32768 h32gqa8", [32768], 32, 8, iters=max(3, iters // 10))
