#!/usr/bin/env python3
"""LLaMA-70B training step on ONE MI355X (288 GB sizing evidence,
config #4 of BASELINE.json): bf16 params+grads resident (~282 GB),
gradient checkpointing, fp32 optimizer states HOST-OFFLOADED and
streamed through the GPU by the chunked async AdamW
(parallel/ddp.py:_adamw_offloaded).  At dp1 the whole 840 GB of fp32
state crosses PCIe both ways per step, so the step is PCIe-bound; at
dp8 each rank moves 1/8th and the same machinery overlaps it with the
bucketed grad reduce-scatter.

Run: python tools/bench_70b_step.py [--steps 2] [--tokens 2048]
     [--model 70b|34b]
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

os.environ.setdefault("PYTORCH_ALLOC_CONF", "expandable_segments:True")

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=2)
    p.add_argument("--tokens", type=int, default=2048)
    p.add_argument("--model", type=str, default="70b")
    p.add_argument("--no-offload", dest="offload", action="store_false",
                   default=True)
    p.add_argument("--seq", type=int, default=512)
    args = p.parse_args()

    from realhf_amd.api.model import ReaLModelConfig
    from realhf_amd.base import constants
    from realhf_amd.models.hf.llama import llama70b_config
    from realhf_amd.models.real_model import ReaLModel
    from realhf_amd.parallel.ddp import OptimizerConfig, ZeRO1Optimizer

    if args.model == "7b":
        from realhf_amd.models.hf.llama import llama7b_config

        cfg = llama7b_config()
    elif args.model == "70b":
        cfg = llama70b_config()
    elif args.model == "34b":  # CodeLlama-34b geometry
        cfg = llama70b_config()
        cfg.n_layers = 48
        cfg.hidden_dim = 8192
        cfg.intermediate_dim = 22016
        cfg.n_heads = 64
        cfg.n_kv_heads = 8
    else:  # 13b geometry (safe-margin offload-tier measurement)
        cfg = llama70b_config()
        cfg.n_layers = 40
        cfg.hidden_dim = 5120
        cfg.intermediate_dim = 13824
        cfg.n_heads = 40
        cfg.n_kv_heads = 40
    print(f"model: {cfg.n_layers}L hidden {cfg.hidden_dim} "
          f"({cfg.param_count()/1e9:.1f}B params)", flush=True)

    t0 = time.time()
    m = ReaLModel(cfg, device="cuda", dtype=torch.bfloat16)
    m.random_init()
    print(f"built in {time.time()-t0:.0f}s; "
          f"mem={torch.cuda.memory_allocated()/2**30:.0f} GiB", flush=True)

    # gradient checkpointing via the topology flag
    from realhf_amd.base.topology import FakeGrid, PipeDataTensorTopology

    topo = PipeDataTensorTopology(num_pp=1, num_dp=1, num_tp=1,
                                  gradient_checkpointing=True)
    constants.set_grid("m70", FakeGrid(0, topo))

    t0 = time.time()
    with constants.model_scope("m70"):
        opt = ZeRO1Optimizer(
            m, OptimizerConfig(lr=1e-5, warmup_steps_proportion=0.0,
                               offload=args.offload),
            total_train_steps=100,
        )
    print(f"optimizer (host fp32 states) in {time.time()-t0:.0f}s; "
          f"dev mem={torch.cuda.memory_allocated()/2**30:.0f} GiB",
          flush=True)

    toks = torch.randint(0, cfg.vocab_size, (args.tokens,), device="cuda")
    seq = args.seq
    cu = torch.arange(0, args.tokens + 1, seq, dtype=torch.int32,
                      device="cuda")

    def step():
        with constants.model_scope("m70"):
            opt.zero_grad()
            out = m(packed_input_ids=toks, cu_seqlens=cu, max_seqlen=seq)
            out.float().square().mean().backward()
            del out
            t_opt = time.time()
            opt.step()
            torch.cuda.synchronize()
            return time.time() - t_opt

    times = []
    for i in range(args.steps):
        torch.cuda.synchronize()
        t0 = time.time()
        t_opt = step()
        dt = time.time() - t0
        times.append(dt)
        print(f"step {i}: {dt:.1f}s total ({t_opt:.1f}s optimizer sweep); "
              f"peak mem={torch.cuda.max_memory_allocated()/2**30:.0f} GiB",
              flush=True)
    flops = 6 * cfg.param_count() * args.tokens
    best = min(times)
    print(f"RESULT {cfg.param_count()/1e9:.1f}B: {best:.1f}s/step "
          f"({args.tokens} tok, grad-ckpt, offloaded optimizer) "
          f"fwd+bwd {flops/1e12:.0f} TFLOP", flush=True)


if __name__ == "__main__":
    main()
