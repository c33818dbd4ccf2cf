"""Tiny attn_fwd workload for PMC collection."""
import os, sys, math
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from dmlcloud_amd import _C
DEV='cuda:0'
b,h,n,d = 64,12,1024,64
q = (torch.randn(b,h,n,d,device=DEV)*0.5).to(torch.bfloat16)
k = (torch.randn(b,h,n,d,device=DEV)*0.5).to(torch.bfloat16)
v = (torch.randn(b,h,n,d,device=DEV)*0.5).to(torch.bfloat16)
o = torch.empty_like(q); lse = torch.empty(b,h,n,dtype=torch.float32,device=DEV)
for _ in range(3):
    _C.attn_fwd(q,k,v,o,lse,1.0/math.sqrt(d),False)
torch.cuda.synchronize(); print('ok')
