"""Flash-attention shape sweep: ours vs torch SDPA across seq lengths
and GQA/MHA head layouts (generality evidence for profiles/)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

import ray_amd.ops as ops


def bench(fn, iters=30, warmup=8):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    torch.manual_seed(0)
    D = 128
    print(f"{'shape':34s} {'ours':>9s} {'SDPA':>9s} {'ratio':>6s}   "
          f"{'ours TF':>8s}")
    for B, Hq, Hkv, T in (
        (16, 32, 8, 2048),
        (8, 32, 8, 4096),
        (2, 32, 8, 8192),
        (8, 32, 32, 4096),   # MHA
        (8, 16, 16, 4096),   # smaller MHA
    ):
        q = torch.randn(B, Hq, T, D, device="cuda", dtype=torch.bfloat16)
        k = torch.randn(B, Hkv, T, D, device="cuda", dtype=torch.bfloat16)
        v = torch.randn(B, Hkv, T, D, device="cuda", dtype=torch.bfloat16)
        t_ours = bench(lambda: ops.flash_attention(q, k, v, causal=True))
        t_sdpa = bench(lambda: F.scaled_dot_product_attention(
            q, k, v, is_causal=True, enable_gqa=True))
        tf = 4 * B * Hq * T * T * D * 0.5 / t_ours / 1e12
        name = f"B{B} Hq{Hq}/Hkv{Hkv} T{T}"
        print(f"{name:34s} {t_ours * 1e3:8.3f}ms {t_sdpa * 1e3:8.3f}ms "
              f"{t_sdpa / t_ours:5.2f}x {tf:8.1f}")



def bwd_sweep():
    torch.manual_seed(0)
    D = 128
    print(f"\nfwd+bwd: {'shape':26s} {'ours':>9s} {'SDPA':>9s} {'ratio':>6s}")
    for B, Hq, Hkv, T in (
        (16, 32, 8, 2048),
        (8, 32, 8, 4096),
        (2, 32, 8, 8192),
        (8, 32, 32, 4096),
    ):
        def mk():
            return (torch.randn(B, Hq, T, D, device="cuda",
                                dtype=torch.bfloat16, requires_grad=True),
                    torch.randn(B, Hkv, T, D, device="cuda",
                                dtype=torch.bfloat16, requires_grad=True),
                    torch.randn(B, Hkv, T, D, device="cuda",
                                dtype=torch.bfloat16, requires_grad=True))
        q, k, v = mk()
        g = torch.randn(B, Hq, T, D, device="cuda", dtype=torch.bfloat16)

        def ours():
            out = ops.flash_attention(q, k, v, causal=True)
            out.backward(g)
            q.grad = k.grad = v.grad = None

        def sdpa():
            out = F.scaled_dot_product_attention(
                q, k, v, is_causal=True, enable_gqa=True)
            out.backward(g)
            q.grad = k.grad = v.grad = None

        t_ours = bench(ours, iters=15, warmup=5)
        t_sdpa = bench(sdpa, iters=15, warmup=5)
        name = f"B{B} Hq{Hq}/Hkv{Hkv} T{T}"
        print(f"fwd+bwd: {name:26s} {t_ours * 1e3:8.2f}ms "
              f"{t_sdpa * 1e3:8.2f}ms {t_sdpa / t_ours:5.2f}x")


if __name__ == "__main__":
    main()
    bwd_sweep()
