#!/usr/bin/env python3
"""PMC counter target: run each flagship MFMA kernel a fixed number of
times at the BENCH shapes (7B PPO: 16 x 640-token sequences, GQA 32/8,
hd 128) so `rocprofv3 --pmc ...` can attribute counters per kernel.

Usage (GPU box):
  cd /tmp && export TMPDIR=/tmp && \
  rocprofv3 --pmc SQ_WAVE_CYCLES,SQ_WAIT_ANY,SQ_WAIT_INST_ANY,\
SQ_LDS_BANK_CONFLICT,SQ_VALU_MFMA_BUSY_CYCLES \
    --kernel-include-regex 'attn_|skinny|grouped' \
    -d OUT -o pmc -- python tools/pmc_probe.py
then tools/pmc_summary.py OUT/.../pmc_counter_collection.csv
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch

import realhf_amd._C as C

ITERS = 10
torch.manual_seed(0)

# ---- flash-attention varlen fwd + bwd, bench shape --------------------
lens = [640] * 16
total = sum(lens)
cu = torch.tensor([0] + list(np.cumsum(lens)), dtype=torch.int32,
                  device="cuda")
nq, nkv, hd = 32, 8, 128
scale = hd ** -0.5
q = (torch.randn(total, nq, hd, device="cuda") * 0.3).to(torch.bfloat16)
k = (torch.randn(total, nkv, hd, device="cuda") * 0.3).to(torch.bfloat16)
v = (torch.randn(total, nkv, hd, device="cuda") * 0.3).to(torch.bfloat16)
out, lse = C.attn_varlen_fwd(q, k, v, cu, max(lens), True, scale, 0)
for _ in range(ITERS):
    out, lse = C.attn_varlen_fwd(q, k, v, cu, max(lens), True, scale, 0)
dout = torch.randn_like(out)
dsum = (dout.float() * out.float()).sum(-1)
for _ in range(ITERS):
    C.attn_varlen_bwd(q, k, v, dout, lse, dsum, cu, True, scale, 0)
torch.cuda.synchronize()

# ---- decode attention, bench decode shape (bs16, ctx 640) -------------
bs, maxlen = 16, 640
dl = torch.full((bs,), 640, dtype=torch.int32, device="cuda")
dq = (torch.randn(bs, nq, hd, device="cuda") * 0.3).to(torch.bfloat16)
kc = (torch.randn(bs, maxlen, nkv, hd, device="cuda") * 0.3).to(torch.bfloat16)
vc = (torch.randn(bs, maxlen, nkv, hd, device="cuda") * 0.3).to(torch.bfloat16)
for _ in range(ITERS):
    C.attn_decode(dq, kc, vc, dl, scale, 0)
torch.cuda.synchronize()

# ---- skinny decode GEMM, 7B qkv/gateup shapes -------------------------
ws = torch.empty(32 * 16 * 32000, dtype=torch.float32, device="cuda")
for N, K, sk in ((12288, 4096, 8), (22016, 4096, 8)):
    x = (torch.randn(16, K, device="cuda") * 0.3).to(torch.bfloat16)
    w = (torch.randn(N, K, device="cuda") * 0.3).to(torch.bfloat16)
    for _ in range(ITERS):
        C.skinny_gemm_nc(x, w, ws, sk)
    torch.cuda.synchronize()

# ---- grouped GEMM (MoE experts, grouped-dispatch regime) --------------
E, N, K = 8, 1792, 2048
glens = torch.tensor([64] * E, dtype=torch.int32)
gx = (torch.randn(int(glens.sum()), K, device="cuda") * 0.3).to(torch.bfloat16)
gw = (torch.randn(E, N, K, device="cuda") * 0.3).to(torch.bfloat16)
for _ in range(ITERS):
    C.grouped_gemm(gx, gw, glens)
torch.cuda.synchronize()
print("pmc probe done")
