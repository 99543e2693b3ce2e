"""Per-op HIP kernel vs torch-eager microbench (evidence table)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F
import ray_amd.ops as ops

dev = "cuda"
def t(fn, n=50):
    for _ in range(10): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / n * 1e6

rows = []
# RMSNorm fwd (llama shape per microbatch)
x = torch.randn(8 * 4096, 4096, device=dev, dtype=torch.bfloat16)
w = torch.ones(4096, device=dev, dtype=torch.bfloat16)
def rms_ref():
    xf = x.float()
    return (xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + 1e-5) * w.float()).to(torch.bfloat16)
with torch.no_grad():
    rows.append(("rmsnorm fwd 32k x 4096", t(lambda: ops.rmsnorm(x, w)), t(rms_ref)))

# SwiGLU fwd
a = torch.randn(8 * 4096, 14336, device=dev, dtype=torch.bfloat16)
b = torch.randn_like(a)
with torch.no_grad():
    rows.append(("swiglu fwd 32k x 14336", t(lambda: ops.swiglu(a, b)),
                 t(lambda: (F.silu(a.float()) * b.float()).to(torch.bfloat16))))

# RoPE fwd
q = torch.randn(8, 4096, 32, 128, device=dev, dtype=torch.bfloat16)
cosT, sinT = ops.rope_tables(4096, 128, device=dev)
with torch.no_grad():
    rows.append(("rope fwd B8 T4096 H32", t(lambda: ops.rope(q, cosT, sinT)),
                 t(lambda: ops.rope_ref(q, cosT, sinT))))

# Fused CE fwd+bwd (logits grad) vs torch CE
logits = torch.randn(8192, 128256, device=dev, dtype=torch.bfloat16, requires_grad=True)
tgt = torch.randint(0, 128256, (8192,), device=dev)
def ce_ours():
    loss = ops.cross_entropy(logits, tgt); loss.backward(); logits.grad = None
def ce_torch():
    loss = F.cross_entropy(logits.float(), tgt); loss.backward(); logits.grad = None
rows.append(("fused CE fwd+bwd 8k x 128k", t(ce_ours, 20), t(ce_torch, 20)))

# Fused AdamW vs torch AdamW (foreach) on 1B params worth of 4096x4096 chunks
ps = [torch.randn(4096, 4096, device=dev, dtype=torch.bfloat16, requires_grad=True) for _ in range(16)]
for p in ps: p.grad = torch.randn_like(p)
opt1 = ops.FusedAdamW(ps, lr=1e-4)
def ours_step(): opt1.step()
ps2 = [torch.randn(4096, 4096, device=dev, dtype=torch.float32, requires_grad=True) for _ in range(16)]
for p in ps2: p.grad = torch.randn_like(p)
opt2 = torch.optim.AdamW(ps2, lr=1e-4, foreach=True)
def torch_step(): opt2.step()
rows.append(("AdamW step 268M params", t(ours_step, 20), t(torch_step, 20)))

print(f"{'op':28s} {'ours us':>9s} {'torch us':>9s} {'speedup':>8s}")
for name, u, v in rows:
    print(f"{name:28s} {u:9.1f} {v:9.1f} {v/u:7.2f}x")
