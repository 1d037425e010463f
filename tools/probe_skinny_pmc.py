#!/usr/bin/env python3
"""Run ONLY the skinny GEMM v1 kernel on the 7B decode shapes, many
iterations, for rocprofv3 --pmc counter collection."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
import realhf_amd._C as C

SHAPES = [("qkv", 12288, 4096, 8), ("o", 4096, 4096, 8),
          ("gateup", 22016, 4096, 8), ("down", 4096, 11008, 16)]
M = 16
ws = torch.empty(32 * M * 32000, dtype=torch.float32, device="cuda")
for name, N, K, sk in SHAPES:
    x = (torch.randn(M, K, device="cuda") * 0.3).to(torch.bfloat16)
    w = (torch.randn(N, K, device="cuda") * 0.3).to(torch.bfloat16)
    for _ in range(30):
        C.skinny_gemm_nc(x, w, ws, sk)
    torch.cuda.synchronize()
print("done")
