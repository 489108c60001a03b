import sys, os
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), '..'))
import torch, os
from megatron_amd import ops
s,b,hq,hkv,d = 4096,4,32,8,128
q = torch.randn(s,b,hq,d, device="cuda", dtype=torch.bfloat16)
k = torch.randn(s,b,hkv,d, device="cuda", dtype=torch.bfloat16)
v = torch.randn(s,b,hkv,d, device="cuda", dtype=torch.bfloat16)
out, lse = ops._C.attn_fwd(q,k,v,True,d**-0.5,0)
dy = torch.randn_like(out)
for _ in range(10):
    ops._C.attn_bwd(dy,q,k,v,out,lse,True,d**-0.5,0)
torch.cuda.synchronize()
