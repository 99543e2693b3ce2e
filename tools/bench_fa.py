"""Quick flash-attention kernel iteration bench (GPU box)."""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import os, time, torch
import torch.nn.functional as F
from ray_amd import ops

def run(tag):
    # correctness small
    torch.manual_seed(0)
    q = torch.randn(2, 8, 256, 128, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(2, 2, 256, 128, device="cuda", dtype=torch.bfloat16)
    v = torch.randn_like(k)
    out, lse = ops.flash_attention(q, k, v, causal=True, return_lse=True)
    ref, lse_ref = ops.flash_attention_ref(q, k, v, causal=True)
    err = (out.float()-ref.float()).abs().max().item()
    lse_err = (lse-lse_ref).abs().max().item()
    # perf big
    B,Hq,Hkv,T,D = 8,32,8,4096,128
    q = torch.randn(B,Hq,T,D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B,Hkv,T,D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn_like(k)
    for _ in range(3): ops.flash_attention(q,k,v,causal=True)
    torch.cuda.synchronize()
    t0=time.perf_counter()
    for _ in range(10): ops.flash_attention(q,k,v,causal=True)
    torch.cuda.synchronize()
    dt=(time.perf_counter()-t0)/10
    flops = 4*B*Hq*T*T*D/2
    print(f"{tag}: err={err:.3f} lse_err={lse_err:.4f}  {dt*1000:.2f} ms  {flops/dt/1e12:.0f} TF")

run("fa-default")
os.environ["RAY_AMD_FA_V2"]="1"
# env is read once (static); need subprocess for A/B — done by caller
# sdpa reference
B,Hq,Hkv,T,D = 8,32,8,4096,128
q = torch.randn(B,Hq,T,D, device="cuda", dtype=torch.bfloat16)
k = torch.randn(B,Hkv,T,D, device="cuda", dtype=torch.bfloat16)
v = torch.randn_like(k)
for _ in range(3): F.scaled_dot_product_attention(q,k,v,is_causal=True,enable_gqa=True)
torch.cuda.synchronize()
t0=time.perf_counter()
for _ in range(10): F.scaled_dot_product_attention(q,k,v,is_causal=True,enable_gqa=True)
torch.cuda.synchronize()
dt=(time.perf_counter()-t0)/10
print(f"sdpa: {dt*1000:.2f} ms  {4*B*Hq*T*T*D/2/dt/1e12:.0f} TF")
