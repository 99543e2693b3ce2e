import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import ray_amd.ops as ops

x = torch.randint(0, 255, (256, 224, 224, 3), dtype=torch.uint8, device="cuda")
mean = torch.tensor([0.485, 0.456, 0.406], device="cuda")
std = torch.tensor([0.229, 0.224, 0.225], device="cuda")

def ref():
    xf = x.float().permute(0, 3, 1, 2) / 255.0
    return ((xf - mean.view(1, 3, 1, 1)) / std.view(1, 3, 1, 1)).to(torch.bfloat16)

def t(fn, n=50):
    for _ in range(10): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / n

a = t(lambda: ops.img_normalize(x, mean, std))
b = t(ref)
gbs = x.numel() * (1 + 2) / a / 1e9  # u8 in + bf16 out
print(f"ours {a*1e6:7.1f} us ({gbs:6.0f} GB/s eff)   torch-eager {b*1e6:7.1f} us   speedup {b/a:4.1f}x")
