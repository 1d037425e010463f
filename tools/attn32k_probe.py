import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch
import realhf_amd._C as C
lens=[32768]; total=sum(lens)
cu=torch.tensor([0]+list(np.cumsum(lens)),dtype=torch.int32,device="cuda")
q=(torch.randn(total,32,128,device="cuda")*0.3).to(torch.bfloat16)
k=(torch.randn(total,8,128,device="cuda")*0.3).to(torch.bfloat16)
v=(torch.randn(total,8,128,device="cuda")*0.3).to(torch.bfloat16)
for _ in range(3):
    C.attn_varlen_fwd(q,k,v,cu,32768,True,0.0883,0)
torch.cuda.synchronize()
