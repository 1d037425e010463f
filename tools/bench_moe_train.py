#!/usr/bin/env python3
"""MoE (Mixtral-style) TRAINING microbench — evidence that the GRPO
Mixtral path trains on the hand-written MFMA grouped-GEMM kernels
(fwd: ops/csrc/grouped_gemm.hip, bwd dX/dW: grouped_gemm_bwd.hip).

Run under rocprofv3 --stats to see grouped_gemm_kernel /
grouped_gemm_dx_kernel / grouped_gemm_dw_kernel among the top entries:

  rocprofv3 --kernel-trace --stats -d gpurun_out/prof_moe -- \
      python tools/bench_moe_train.py --steps 5
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=5)
    p.add_argument("--warmup", type=int, default=2)
    p.add_argument("--layers", type=int, default=4)
    p.add_argument("--hidden", type=int, default=2048)
    p.add_argument("--inter", type=int, default=7168)
    p.add_argument("--experts", type=int, default=8)
    p.add_argument("--tokens", type=int, default=8192)
    p.add_argument("--loop", action="store_true",
                   help="force the per-expert rocBLAS loop (A/B baseline)")
    p.add_argument("--grouped", action="store_true",
                   help="force the MFMA grouped kernel at any segment size")
    args = p.parse_args()
    if args.loop:
        os.environ["REALHF_AMD_MOE_LOOP"] = "1"
    if args.grouped:
        os.environ["REALHF_AMD_MOE_GROUPED"] = "1"

    import realhf_amd.models.hf as hf_reg
    from realhf_amd.models.real_model import ReaLModel

    fam = hf_reg.get_family("mixtral")
    cfg = fam.make_test_config(
        n_layers=args.layers, hidden_dim=args.hidden, n_heads=16,
        n_kv_heads=8, head_dim=128, intermediate_dim=args.inter,
        vocab_size=32000,
    )
    cfg.moe.num_experts = args.experts
    torch.manual_seed(7)
    m = ReaLModel(cfg, device="cuda", dtype=torch.bfloat16)
    m.random_init()
    m.train()
    m.allocate_grad_buffer()
    for k, pm in m._params.items():
        pm.requires_grad_(True)
        pm.grad = m.grad_view(k)

    bs = 16
    seq = args.tokens // bs
    toks = torch.randint(0, 32000, (args.tokens,), device="cuda")
    cu = torch.arange(0, args.tokens + 1, seq, dtype=torch.int32,
                      device="cuda")

    def step():
        m.flat_grad.zero_()
        out = m(packed_input_ids=toks, cu_seqlens=cu, max_seqlen=seq)
        out.float().square().mean().backward()

    for _ in range(args.warmup):
        step()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(args.steps):
        step()
    torch.cuda.synchronize()
    dt = (time.time() - t0) / args.steps
    # fwd+bwd expert GEMM FLOPs: 3 GEMMs x 2 (topk) x 3 (fwd+bwd)
    moe_flops = 3 * 2 * args.tokens * 2 * 3 * args.hidden * args.inter \
        * args.layers
    print(f"moe train step: {dt*1e3:.1f} ms/step "
          f"({args.tokens} tok, {args.experts} experts x {args.layers} L, "
          f"hidden {args.hidden}, inter {args.inter}); "
          f"expert-GEMM {moe_flops/dt/1e12:.0f} TFLOP/s "
          f"{'(per-expert loop)' if args.loop else ('(grouped MFMA)' if args.grouped else '(adaptive)')}")


if __name__ == "__main__":
    main()
