#!/usr/bin/env python3
"""Measure the machine constants the allocation-search cost model uses
(reference counterpart: search_engine/layers.py ProfileLayers — runs real
ops and dumps cost tables).  Run on an MI355X:

    python tools/profile_layers.py [-o cost_table.json]

then point the search at it:  REALHF_AMD_COST_TABLE=cost_table.json
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import torch


def _time(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("-o", "--out", default="cost_table.json")
    args = ap.parse_args()
    assert torch.cuda.is_available(), "profile_layers needs a GPU"

    # dense bf16 GEMM at a training-ish shape (tokens x hidden x 4h)
    m, k, n = 8192, 4096, 11008
    a = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(n, k, device="cuda", dtype=torch.bfloat16)
    t = _time(lambda: a @ b.t())
    bf16_tf = 2 * m * k * n / t / 1e12

    # HBM stream (copy reads+writes)
    x = torch.empty(1 << 28, device="cuda", dtype=torch.bfloat16)  # 512 MB
    y = torch.empty_like(x)
    t = _time(lambda: y.copy_(x))
    hbm_gbps = 2 * x.numel() * 2 / t / 1e9

    table = {"bf16_tf": round(bf16_tf, 1), "hbm_gbps": round(hbm_gbps, 1)}

    # xGMI: all-reduce bus bandwidth when launched under torchrun
    if int(os.environ.get("WORLD_SIZE", "1")) > 1:
        import torch.distributed as dist

        dist.init_process_group("nccl")
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", 0)))
        z = torch.empty(1 << 27, device="cuda", dtype=torch.bfloat16)
        t = _time(lambda: dist.all_reduce(z))
        w = dist.get_world_size()
        bus = 2 * (w - 1) / w * z.numel() * 2 / t / 1e9
        table["xgmi_link_gbps"] = round(bus / 7, 1)  # 7 links per GPU
        dist.destroy_process_group()

    # end-to-end efficiencies on a real (small) llama: these drive the
    # search engine's per-MFC time estimates (reference counterpart:
    # profiled per-op tables, search_engine/estimate.py:135-450)
    from realhf_amd.api.model import GenerationHyperparameters
    from realhf_amd.models.generation import generate
    from realhf_amd.models.hf.llama import make_test_config
    from realhf_amd.models.real_model import ReaLModel

    cfg = make_test_config(
        n_layers=16, hidden_dim=2048, n_heads=16, n_kv_heads=16,
        head_dim=128, intermediate_dim=5632, vocab_size=32000,
    )
    cfg.family = "llama"
    model = ReaLModel(cfg, device="cuda", dtype=torch.bfloat16)
    model.random_init()
    p = cfg.param_count()
    tokens = 4096
    toks = torch.randint(0, 32000, (tokens,), device="cuda")
    cu = torch.arange(0, tokens + 1, 512, dtype=torch.int32, device="cuda")
    fwd_flops = 2.0 * p * tokens

    with torch.no_grad():
        t = _time(lambda: model(packed_input_ids=toks, cu_seqlens=cu,
                                max_seqlen=512), iters=5, warmup=2)
    table["inf_eff"] = round(fwd_flops / t / 1e12 / table["bf16_tf"], 3)

    model.allocate_grad_buffer()
    for kk, pm in model._params.items():
        pm.requires_grad_(True)
        pm.grad = model.grad_view(kk)

    def _train():
        model.flat_grad.zero_()
        out = model(packed_input_ids=toks, cu_seqlens=cu, max_seqlen=512)
        out.float().square().mean().backward()

    t = _time(_train, iters=5, warmup=2)
    table["train_eff"] = round(3 * fwd_flops / t / 1e12 / table["bf16_tf"], 3)

    # decode: per-token time -> fraction of HBM weight-streaming bound
    for kk, pm in model._params.items():
        pm.requires_grad_(False)
        pm.grad = None
    bs = 64
    ptoks = torch.randint(0, 32000, (bs * 64,), device="cuda")
    pcu = torch.arange(0, bs * 64 + 1, 64, dtype=torch.int32, device="cuda")
    g = GenerationHyperparameters(max_new_tokens=64, min_new_tokens=64,
                                  greedy=True, use_hip_graph=True)
    with torch.no_grad():
        generate(model, ptoks, pcu, g)  # warm session + graph
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        generate(model, ptoks, pcu, g)
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
    per_tok = dt / 64
    ideal = (p * 2) / (table["hbm_gbps"] * 1e9)
    table["gen_bw_eff"] = round(ideal / per_tok, 3)

    with open(args.out, "w") as f:
        json.dump(table, f, indent=1)
    print(json.dumps(table))


if __name__ == "__main__":
    main()
