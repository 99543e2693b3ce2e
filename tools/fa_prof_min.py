"""Minimal fwd+bwd loop for rocprof (keep trace small)."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import ray_amd.ops as ops

B, Hq, Hkv, T, D = 8, 32, 8, 4096, 128
q = torch.randn(B, Hq, T, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
k = torch.randn(B, Hkv, T, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
v = torch.randn_like(k).requires_grad_()
g = torch.randn(B, Hq, T, D, device="cuda", dtype=torch.bfloat16)
for _ in range(5):
    out = ops.flash_attention(q, k, v, causal=True)
    out.backward(g)
    q.grad = k.grad = v.grad = None
torch.cuda.synchronize()
print("done")
