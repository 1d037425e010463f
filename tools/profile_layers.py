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

    with open(args.out, "w") as f:
        json.dump(table, f, indent=1)
    print(json.dumps(table))


if __name__ == "__main__":
    main()
