"""Flash-attention forward throughput on MI355X (evidence for profiles/).

Measures ray_amd.ops.flash_attention (HIP v5 kernel) vs torch SDPA on
the Llama-3-8B prefill shape, plus max error vs an fp32 reference.
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import ray_amd.ops as ops


def bench(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    assert torch.cuda.is_available()
    torch.manual_seed(0)
    B, H, T, D = 8, 32, 4096, 128
    causal = True
    q = torch.randn(B, H, T, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, H, T, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, H, T, D, device="cuda", dtype=torch.bfloat16)

    # causal flops: ~half the full score matrix
    flops = 4 * B * H * T * T * D * (0.5 if causal else 1.0)

    t_ours = bench(lambda: ops.flash_attention(q, k, v, causal=causal))
    t_sdpa = bench(
        lambda: torch.nn.functional.scaled_dot_product_attention(
            q, k, v, is_causal=causal
        )
    )

    o = ops.flash_attention(q, k, v, causal=causal)
    ref = torch.nn.functional.scaled_dot_product_attention(
        q.float(), k.float(), v.float(), is_causal=causal
    )
    err = (o.float() - ref).abs().max().item()

    print(
        f"shape B{B} H{H} T{T} D{D} causal={causal}\n"
        f"ray_amd flash    : {t_ours*1e3:.3f} ms  "
        f"{flops/t_ours/1e12:.1f} TFLOP/s\n"
        f"torch SDPA      : {t_sdpa*1e3:.3f} ms  "
        f"{flops/t_sdpa/1e12:.1f} TFLOP/s\n"
        f"max |err| vs fp32 ref: {err:.4f}"
    )

    # ---- backward (fwd+bwd round trip, grads on) ----
    bwd_flops = flops * 3.5  # fwd (1x) + bwd (2.5x)
    qg = q.clone().requires_grad_()
    kg = k.clone().requires_grad_()
    vg = v.clone().requires_grad_()
    g = torch.randn_like(q)

    def ours_fb():
        out = ops.flash_attention(qg, kg, vg, causal=causal)
        out.backward(g)
        qg.grad = kg.grad = vg.grad = None

    def sdpa_fb():
        out = torch.nn.functional.scaled_dot_product_attention(
            qg, kg, vg, is_causal=causal
        )
        out.backward(g)
        qg.grad = kg.grad = vg.grad = None

    t_ours_fb = bench(ours_fb, iters=20, warmup=5)
    t_sdpa_fb = bench(sdpa_fb, iters=20, warmup=5)
    print(
        f"fwd+bwd ray_amd : {t_ours_fb*1e3:.3f} ms  "
        f"{bwd_flops/t_ours_fb/1e12:.1f} TFLOP/s\n"
        f"fwd+bwd SDPA    : {t_sdpa_fb*1e3:.3f} ms  "
        f"{bwd_flops/t_sdpa_fb/1e12:.1f} TFLOP/s"
    )
    return 0


if __name__ == "__main__":
    sys.exit(main())
