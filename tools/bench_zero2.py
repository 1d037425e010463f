#!/usr/bin/env python3
"""ZeRO-1 vs ZeRO-2 A/B on a multi-GPU node: peak grad memory and step
time per stage (parallel/ddp.py zero_stage).  ZeRO-2 never allocates the
full-model grad buffer — grads stage through pooled bucket buffers and
reduce-scatter per microbatch — trading n_mbs x RS traffic for
~param-bytes of memory.

Launch (dp over all visible GPUs):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
      --master-addr 127.0.0.1 tools/bench_zero2.py \
      [--model 7b] [--tokens 4096] [--mbs 4] [--steps 3]

Single GPU falls back to gloo-free dp1 where ZeRO-2 disables itself —
this A/B needs dp>1 to say anything.
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch
import torch.distributed as dist


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", type=str, default="7b", choices=["tiny", "7b", "13b"])
    p.add_argument("--tokens", type=int, default=4096)
    p.add_argument("--mbs", type=int, default=4)
    p.add_argument("--steps", type=int, default=3)
    args = p.parse_args()

    from realhf_amd.base import constants
    from realhf_amd.base.testing import init_global_constants
    from realhf_amd.models.hf.llama import make_test_config
    from realhf_amd.models.real_model import ReaLModel
    from realhf_amd.parallel.ddp import OptimizerConfig, ZeRO1Optimizer

    use_cuda = torch.cuda.is_available()
    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    if world > 1:
        dist.init_process_group("nccl" if use_cuda else "gloo",
                                rank=rank, world_size=world)
    if use_cuda:
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", 0)))
    dev = "cuda" if use_cuda else "cpu"
    dtype = torch.bfloat16 if use_cuda else torch.float32

    shapes = {
        "tiny": dict(n_layers=4, hidden_dim=512, n_heads=8, n_kv_heads=8,
                     vocab_size=32000, intermediate_dim=1408),
        "7b": dict(n_layers=32, hidden_dim=4096, n_heads=32, n_kv_heads=32,
                   vocab_size=32000, intermediate_dim=11008,
                   max_position_embeddings=4096),
        "13b": dict(n_layers=40, hidden_dim=5120, n_heads=40, n_kv_heads=40,
                    vocab_size=32000, intermediate_dim=13824,
                    max_position_embeddings=4096),
    }[args.model]

    for stage in (1, 2):
        init_global_constants(num_dp=world, num_tp=1, num_pp=1,
                              model_name=f"m{stage}")
        cfg = make_test_config(**shapes)
        cfg.dtype = str(dtype).split(".")[-1]
        cfg.gradient_checkpointing = args.model != "tiny"
        with constants.model_scope(f"m{stage}"):
            m = ReaLModel(cfg, device=dev, dtype=dtype)
            m.random_init()
            opt = ZeRO1Optimizer(
                m, OptimizerConfig(lr=1e-4, warmup_steps_proportion=0.0,
                                   zero_stage=stage))
            if stage == 2 and world > 1:
                assert opt.zero2, "ZeRO-2 did not engage (needs dp>1 tp1 pp1)"
            rng = np.random.RandomState(7 + rank)
            seq = 512
            n_seq = max(1, args.tokens // seq)
            if use_cuda:
                torch.cuda.reset_peak_memory_stats()
            times = []
            for it in range(args.steps + 1):
                t0 = time.time()
                opt.zero_grad()
                for i in range(args.mbs):
                    toks = torch.from_numpy(
                        rng.randint(0, cfg.vocab_size, size=n_seq * seq)
                    ).long().to(dev)
                    cu = torch.arange(0, n_seq * seq + 1, seq,
                                      dtype=torch.int32, device=dev)
                    if i == args.mbs - 1:
                        opt.arm_overlap()
                    out = m(packed_input_ids=toks, cu_seqlens=cu,
                            max_seqlen=seq)
                    (out.float().square().mean() / args.mbs).backward()
                    opt.end_microbatch()
                opt.step()
                if use_cuda:
                    torch.cuda.synchronize()
                if it > 0:  # first step warms up
                    times.append(time.time() - t0)
            peak = (torch.cuda.max_memory_allocated() / 2**30
                    if use_cuda else 0.0)
            if rank == 0:
                print(f"zero_stage={stage}: {np.mean(times):.3f}s/step "
                      f"(+-{np.std(times):.3f}), peak {peak:.1f} GiB, "
                      f"params {m.flat_param.numel() / 1e9:.2f}B")
            del m, opt
            if use_cuda:
                torch.cuda.empty_cache()
        constants.clear_grids()


if __name__ == "__main__":
    main()
