import os, sys, time
import torch

def t(fn, iters=30, warm=5):
    for _ in range(warm): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/iters

shapes = [(8192,4096,11008), (16384,8192,8192), (2560,4096,11008),
          (2560,11008,4096), (2560,4096,12288)]
for m,k,n in shapes:
    a = torch.randn(m,k,device="cuda",dtype=torch.bfloat16)
    b = torch.randn(n,k,device="cuda",dtype=torch.bfloat16)
    dt = t(lambda: a @ b.t())
    print(f"{m}x{k}x{n}: {2*m*k*n/dt/1e12:.0f} TF/s", flush=True)
