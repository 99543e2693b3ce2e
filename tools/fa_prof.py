import torch
from ray_amd import ops
B,Hq,Hkv,T,D = 8,32,8,4096,128
q = torch.randn(B,Hq,T,D, device="cuda", dtype=torch.bfloat16)
k = torch.randn(B,Hkv,T,D, device="cuda", dtype=torch.bfloat16)
v = torch.randn_like(k)
for _ in range(3): ops.flash_attention(q,k,v,causal=True)
torch.cuda.synchronize()
