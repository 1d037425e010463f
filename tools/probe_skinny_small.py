import sys, os
sys.path.insert(0, "/root/repo")
import torch
import realhf_amd._C as C
M = 16
ws = torch.empty(32 * M * 22016, dtype=torch.float32, device="cuda")
for N, K, sk in ((12288, 4096, 8), (4096, 4096, 8)):
    x = (torch.randn(M, K, device="cuda") * 0.3).to(torch.bfloat16)
    w = (torch.randn(N, K, device="cuda") * 0.3).to(torch.bfloat16)
    for _ in range(20):
        C.skinny_gemm_nc(x, w, ws, sk)
    torch.cuda.synchronize()
print("ok")
