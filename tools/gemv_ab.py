import time, torch, sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from ray_amd import ops
dev = "cuda"
def t(fn, n=50):
    for _ in range(10): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / n * 1e6
for B in (1, 2, 4, 8):
    for N, K in ((4096, 4096), (6144, 4096), (14336, 4096), (4096, 14336), (128256, 4096)):
        x = torch.randn(B, K, device=dev, dtype=torch.bfloat16)
        w = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
        with torch.no_grad():
            us_g = t(lambda: ops._K.gemv(x, w))
            us_l = t(lambda: torch.nn.functional.linear(x, w))
        bw = N * K * 2 / (us_g * 1e-6) / 1e12
        print(f"B{B} N{N:6d} K{K:5d}: gemv {us_g:7.1f}us ({bw:5.2f} TB/s W-read)  linear {us_l:7.1f}us  speedup {us_l/us_g:4.2f}x")
