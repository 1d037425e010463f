#!/usr/bin/env python3
"""Decode attention A/B at llama-7B geometry: us and effective TB/s
per context length.  REALHF_AMD_DEC_WAVES selects the wave-count variant
(the kernel reads it once per process)."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
import realhf_amd._C as C


def bench(fn, iters=100):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    bs, nq, nkv, hd, maxlen = 16, 32, 32, 128, 704
    torch.manual_seed(0)
    q = (torch.randn(bs, nq, hd, device="cuda") * 0.3).to(torch.bfloat16)
    kc = (torch.randn(bs, maxlen, nkv, hd, device="cuda") * 0.3).to(torch.bfloat16)
    vc = (torch.randn(bs, maxlen, nkv, hd, device="cuda") * 0.3).to(torch.bfloat16)
    scale = hd ** -0.5
    print(f"waves={os.environ.get('REALHF_AMD_DEC_WAVES', '4(default)')}")
    print(f"{'ctx':>5} {'us':>8} {'TB/s':>6}")
    for ctx in (128, 256, 384, 512, 640):
        cs = torch.full((bs,), ctx, dtype=torch.int32, device="cuda")
        t = bench(lambda: C.attn_decode(q, kc, vc, cs, scale, 0))
        bytes_ = bs * nkv * ctx * hd * 2 * 2
        print(f"{ctx:5d} {t:8.1f} {bytes_ / t / 1e6:6.2f}")


if __name__ == "__main__":
    main()
