import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import ray_amd.ops as ops

dev = "cuda"
T, B = 512, 4096  # big rollout
r = torch.randn(T, B, device=dev)
v = torch.randn(T + 1, B, device=dev)
c = torch.ones(T, B, device=dev)

def gae_ref():
    adv = torch.zeros(T, B, device=dev)
    last = torch.zeros(B, device=dev)
    for t in range(T - 1, -1, -1):
        delta = r[t] + 0.99 * v[t + 1] * c[t] - v[t]
        last = delta + 0.99 * 0.95 * c[t] * last
        adv[t] = last
    return adv

def t_(fn, n=20):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / n * 1e3

a = t_(lambda: ops.gae(r, v, c, 0.99, 0.95))
b = t_(gae_ref, 5)
print(f"GAE scan T{T} B{B}: ours {a:7.3f} ms   sequential-torch {b:8.1f} ms   {b/a:6.0f}x")

rho = torch.rand(T, B, device=dev)
def vt():
    return ops.vtrace(r, v, c, rho, 0.99, 1.0, 1.0)
a2 = t_(vt)
print(f"V-trace scan T{T} B{B}: ours {a2:7.3f} ms")
