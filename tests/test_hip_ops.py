"""Numerics tests for the CDNA4 HIP kernels vs plain PyTorch fp32
references (SURVEY.md §4 test strategy)."""
import numpy as np
import pytest
import torch

from ray_amd import ops

GPU = pytest.mark.gpu


def _cuda():
    if not torch.cuda.is_available():
        pytest.skip("needs GPU")
    return torch.device("cuda", 0)


# ---------------- CPU reference self-checks (run everywhere) ----------------


def test_gae_ref_matches_manual():
    T, B = 5, 3
    torch.manual_seed(0)
    r = torch.randn(T, B)
    v = torch.randn(T + 1, B)
    cont = torch.ones(T, B)
    adv, vt = ops.gae_ref(r, v, cont, 0.9, 0.8)
    # manual per-env scan
    for b in range(B):
        running = 0.0
        for t in reversed(range(T)):
            delta = r[t, b] + 0.9 * v[t + 1, b] - v[t, b]
            running = float(delta + 0.9 * 0.8 * running)
            assert abs(adv[t, b].item() - running) < 1e-5


def test_vtrace_ref_no_offpolicy_reduces_to_returns():
    # With log_rhos=0 and clips=1, vs should equal the lambda=1 returns.
    T, B = 6, 2
    torch.manual_seed(1)
    r = torch.rand(T, B)
    v = torch.zeros(T + 1, B)
    cont = torch.ones(T, B)
    vs, pg = ops.vtrace_ref(torch.zeros(T, B), r, v, cont, 0.9)
    expected = torch.zeros(B)
    for t in reversed(range(T)):
        expected = r[t] + 0.9 * expected
        if t == 0:
            np.testing.assert_allclose(vs[0].numpy(), expected.numpy(), rtol=1e-5)


# ---------------- GPU numerics ----------------


@GPU
def test_rmsnorm_fwd_bwd():
    dev = _cuda()
    torch.manual_seed(0)
    x = torch.randn(64, 512, device=dev, dtype=torch.bfloat16, requires_grad=True)
    w = torch.randn(512, device=dev, dtype=torch.bfloat16, requires_grad=True)
    y = ops.rmsnorm(x, w)

    xr = x.detach().float().clone().requires_grad_(True)
    wr = w.detach().float().clone().requires_grad_(True)
    inv = torch.rsqrt(xr.pow(2).mean(-1, keepdim=True) + 1e-5)
    yr = xr * inv * wr

    assert torch.allclose(y.float(), yr, atol=3e-2, rtol=3e-2)
    g = torch.randn_like(yr)
    y.backward(g.bfloat16())
    yr.backward(g)
    assert torch.allclose(x.grad.float(), xr.grad, atol=5e-2, rtol=5e-2)
    assert torch.allclose(
        w.grad.float(), wr.grad, atol=8e-1, rtol=5e-2
    )  # dw sums over 64 rows of bf16 products


@GPU
def test_swiglu_fwd_bwd():
    dev = _cuda()
    torch.manual_seed(0)
    a = torch.randn(1024, 64, device=dev, dtype=torch.bfloat16, requires_grad=True)
    b = torch.randn(1024, 64, device=dev, dtype=torch.bfloat16, requires_grad=True)
    y = ops.swiglu(a, b)
    ar = a.detach().float().clone().requires_grad_(True)
    br = b.detach().float().clone().requires_grad_(True)
    yr = torch.nn.functional.silu(ar) * br
    assert torch.allclose(y.float(), yr, atol=3e-2, rtol=3e-2)
    g = torch.randn_like(yr)
    y.backward(g.bfloat16())
    yr.backward(g)
    assert torch.allclose(a.grad.float(), ar.grad, atol=5e-2, rtol=5e-2)
    assert torch.allclose(b.grad.float(), br.grad, atol=5e-2, rtol=5e-2)


@GPU
def test_rope_matches_reference():
    dev = _cuda()
    torch.manual_seed(0)
    B, T, Hn, D = 2, 16, 4, 64
    cosT, sinT = ops.rope_tables(T, D, device=dev)
    x = torch.randn(B, T, Hn, D, device=dev, dtype=torch.bfloat16)
    y = ops.rope(x, cosT, sinT)
    yr = ops.rope_ref(x, cosT, sinT)
    assert torch.allclose(y.float(), yr.float(), atol=2e-2, rtol=2e-2)


@GPU
def test_rope_bwd_is_inverse_rotation():
    dev = _cuda()
    B, T, Hn, D = 1, 8, 2, 32
    cosT, sinT = ops.rope_tables(T, D, device=dev)
    x = torch.randn(B, T, Hn, D, device=dev, dtype=torch.bfloat16, requires_grad=True)
    y = ops.rope(x, cosT, sinT)
    g = torch.randn_like(y)
    y.backward(g)
    # rotation is orthogonal: grad = rotate(g, -theta); |grad| == |g|
    assert torch.allclose(
        x.grad.float().norm(), g.float().norm(), rtol=2e-2
    )


@GPU
def test_cross_entropy_fused():
    dev = _cuda()
    torch.manual_seed(0)
    N, V = 128, 1000
    logits = torch.randn(N, V, device=dev, dtype=torch.bfloat16, requires_grad=True)
    targets = torch.randint(0, V, (N,), device=dev)
    loss = ops.cross_entropy(logits, targets)
    lr = logits.detach().float().clone().requires_grad_(True)
    loss_ref = torch.nn.functional.cross_entropy(lr, targets)
    assert abs(float(loss) - float(loss_ref)) < 3e-2
    loss.backward()
    loss_ref.backward()
    assert torch.allclose(logits.grad.float(), lr.grad, atol=1e-3, rtol=5e-2)


@GPU
def test_cross_entropy_ignore_index():
    dev = _cuda()
    N, V = 64, 512
    logits = torch.randn(N, V, device=dev, dtype=torch.bfloat16)
    targets = torch.randint(0, V, (N,), device=dev)
    targets[::2] = -100
    loss = ops.cross_entropy(logits, targets)
    loss_ref = torch.nn.functional.cross_entropy(
        logits.detach().float(), targets, ignore_index=-100
    )
    assert abs(float(loss) - float(loss_ref)) < 5e-2


@GPU
def test_adamw_matches_torch():
    dev = _cuda()
    torch.manual_seed(0)
    p0 = torch.randn(1000, device=dev, dtype=torch.float32)
    g = torch.randn(1000, device=dev, dtype=torch.float32)

    p_ours = p0.clone().requires_grad_(True)
    p_ours.grad = g.clone()
    opt = ops.FusedAdamW([p_ours], lr=1e-2, betas=(0.9, 0.99), eps=1e-8,
                         weight_decay=0.01)

    p_ref = p0.clone().requires_grad_(True)
    p_ref.grad = g.clone()
    ref = torch.optim.AdamW([p_ref], lr=1e-2, betas=(0.9, 0.99), eps=1e-8,
                            weight_decay=0.01)
    for _ in range(5):
        opt.step()
        ref.step()
    assert torch.allclose(p_ours, p_ref, atol=1e-4, rtol=1e-4)


@GPU
def test_adamw_bf16_master():
    dev = _cuda()
    torch.manual_seed(0)
    p = torch.randn(4096, device=dev, dtype=torch.bfloat16).requires_grad_(True)
    p.grad = torch.randn_like(p)
    opt = ops.FusedAdamW([p], lr=1e-2)
    before = p.detach().float().clone()
    opt.step()
    assert not torch.allclose(p.detach().float(), before)


@GPU
def test_gae_gpu_matches_ref():
    dev = _cuda()
    torch.manual_seed(0)
    T, B = 128, 512
    r = torch.randn(T, B, device=dev)
    v = torch.randn(T + 1, B, device=dev)
    cont = (torch.rand(T, B, device=dev) > 0.05).float()
    adv, vt = ops.gae(r, v, cont, 0.99, 0.95)
    adv_ref, vt_ref = ops.gae_ref(r.cpu(), v.cpu(), cont.cpu(), 0.99, 0.95)
    assert torch.allclose(adv.cpu(), adv_ref, atol=1e-4, rtol=1e-4)
    assert torch.allclose(vt.cpu(), vt_ref, atol=1e-4, rtol=1e-4)


@GPU
def test_vtrace_gpu_matches_ref():
    dev = _cuda()
    torch.manual_seed(0)
    T, B = 64, 256
    lr_ = 0.3 * torch.randn(T, B, device=dev)
    r = torch.randn(T, B, device=dev)
    v = torch.randn(T + 1, B, device=dev)
    cont = (torch.rand(T, B, device=dev) > 0.05).float()
    vs, pg = ops.vtrace(lr_, r, v, cont, 0.99)
    vs_ref, pg_ref = ops.vtrace_ref(lr_.cpu(), r.cpu(), v.cpu(), cont.cpu(), 0.99)
    assert torch.allclose(vs.cpu(), vs_ref, atol=1e-4, rtol=1e-3)
    assert torch.allclose(pg.cpu(), pg_ref, atol=1e-4, rtol=1e-3)


@GPU
def test_img_normalize():
    dev = _cuda()
    x = torch.randint(0, 256, (4, 32, 32, 3), device=dev, dtype=torch.uint8)
    mean = torch.tensor([0.485, 0.456, 0.406], device=dev)
    std = torch.tensor([0.229, 0.224, 0.225], device=dev)
    y = ops.img_normalize(x, mean, std)
    yr = ops.img_normalize_ref(x, mean, std)
    assert y.shape == (4, 3, 32, 32)
    assert torch.allclose(y.float(), yr.float(), atol=2e-2, rtol=2e-2)


@GPU
def test_tiny_llama_step_on_gpu():
    dev = _cuda()
    from ray_amd.models.llama import CONFIGS, LlamaModel

    cfg = CONFIGS["llama-tiny"]
    m = LlamaModel(cfg, dtype=torch.bfloat16).to(dev)
    m.cosT = m.cosT.to(dev)
    m.sinT = m.sinT.to(dev)
    tokens = torch.randint(0, cfg.vocab_size, (2, 64), device=dev)
    targets = torch.randint(0, cfg.vocab_size, (2, 64), device=dev)
    opt = ops.FusedAdamW(m.parameters(), lr=1e-3)
    losses = []
    for _ in range(8):
        loss = m(tokens, targets)
        loss.backward()
        opt.step()
        opt.zero_grad()
        losses.append(float(loss.detach().float().cpu()))
    assert losses[-1] < losses[0], losses


@GPU
def test_flash_attention_fwd_causal():
    dev = _cuda()
    torch.manual_seed(0)
    B, Hq, Hkv, T, D = 2, 8, 2, 256, 128
    q = torch.randn(B, Hq, T, D, device=dev, dtype=torch.bfloat16)
    k = torch.randn(B, Hkv, T, D, device=dev, dtype=torch.bfloat16)
    v = torch.randn(B, Hkv, T, D, device=dev, dtype=torch.bfloat16)
    out, lse = ops.flash_attention(q, k, v, causal=True, return_lse=True)
    ref, lse_ref = ops.flash_attention_ref(q, k, v, causal=True)
    assert torch.allclose(out.float(), ref.float(), atol=4e-2, rtol=4e-2), (
        (out.float() - ref.float()).abs().max()
    )
    assert torch.allclose(lse, lse_ref, atol=2e-2, rtol=2e-2)


@GPU
def test_flash_attention_fwd_full():
    dev = _cuda()
    torch.manual_seed(1)
    B, Hq, Hkv, T, D = 1, 4, 4, 192, 128
    q = torch.randn(B, Hq, T, D, device=dev, dtype=torch.bfloat16)
    k = torch.randn(B, Hkv, T, D, device=dev, dtype=torch.bfloat16)
    v = torch.randn(B, Hkv, T, D, device=dev, dtype=torch.bfloat16)
    out = ops.flash_attention(q, k, v, causal=False)
    ref, _ = ops.flash_attention_ref(q, k, v, causal=False)
    assert torch.allclose(out.float(), ref.float(), atol=4e-2, rtol=4e-2)


@GPU
def test_flash_attention_vs_sdpa_speed():
    import time

    import torch.nn.functional as F

    dev = _cuda()
    B, Hq, Hkv, T, D = 8, 32, 8, 4096, 128
    q = torch.randn(B, Hq, T, D, device=dev, dtype=torch.bfloat16)
    k = torch.randn(B, Hkv, T, D, device=dev, dtype=torch.bfloat16)
    v = torch.randn(B, Hkv, T, D, device=dev, dtype=torch.bfloat16)

    def ours():
        return ops.flash_attention(q, k, v, causal=True)

    def sdpa():
        return F.scaled_dot_product_attention(q, k, v, is_causal=True,
                                              enable_gqa=True)

    for f in (ours, sdpa):
        f()
    torch.cuda.synchronize()
    times = {}
    for name, f in (("ours", ours), ("sdpa", sdpa)):
        t0 = time.perf_counter()
        for _ in range(5):
            f()
        torch.cuda.synchronize()
        times[name] = (time.perf_counter() - t0) / 5
    flops = 4 * B * Hq * T * T * D / 2
    print({k: f"{v*1000:.1f}ms ({flops/v/1e12:.0f} TF)" for k, v in times.items()})
    # correctness at this scale too (sampled)
    o1 = ours().float()
    o2 = sdpa().float()
    assert torch.allclose(o1, o2, atol=5e-2, rtol=5e-2)


def test_flash_attention_cpu_backward():
    """CPU path of flash_attention is differentiable via the fp32 ref."""
    q = torch.randn(1, 2, 64, 128, requires_grad=True)
    k = torch.randn(1, 2, 64, 128, requires_grad=True)
    v = torch.randn(1, 2, 64, 128, requires_grad=True)
    out = ops.flash_attention(q, k, v, causal=True)
    out.sum().backward()
    assert q.grad is not None and torch.isfinite(q.grad).all()


@pytest.mark.gpu
def test_flash_attention_backward_gpu():
    """HIP fa_bwd (dq + dkv kernels) vs fp32 autograd of the reference,
    over causal/non-causal and GQA shapes."""
    torch.manual_seed(0)
    for B, Hq, Hkv, T, causal in [
        (2, 4, 2, 256, True),
        (1, 8, 8, 128, False),
        (1, 32, 8, 512, True),
    ]:
        q = torch.randn(B, Hq, T, 128, device="cuda",
                        dtype=torch.bfloat16, requires_grad=True)
        k = torch.randn(B, Hkv, T, 128, device="cuda",
                        dtype=torch.bfloat16, requires_grad=True)
        v = torch.randn(B, Hkv, T, 128, device="cuda",
                        dtype=torch.bfloat16, requires_grad=True)
        g = torch.randn(B, Hq, T, 128, device="cuda",
                        dtype=torch.bfloat16)

        out = ops.flash_attention(q, k, v, causal=causal)
        out.backward(g)
        dq, dk, dv = q.grad.clone(), k.grad.clone(), v.grad.clone()

        q2 = q.detach().clone().requires_grad_()
        k2 = k.detach().clone().requires_grad_()
        v2 = v.detach().clone().requires_grad_()
        ref, _ = ops.flash_attention_ref(q2, k2, v2, causal=causal)
        ref.backward(g)

        for name, ours, refg in (
            ("dq", dq, q2.grad), ("dk", dk, k2.grad), ("dv", dv, v2.grad)
        ):
            ours = ours.float()
            refg = refg.float()
            rel = (ours - refg).norm() / (refg.norm() + 1e-6)
            assert torch.isfinite(ours).all(), (name, B, Hq, T, causal)
            assert rel < 0.03, (name, B, Hq, Hkv, T, causal, float(rel))


@pytest.mark.gpu
def test_flash_attention_bthd_layout_matches():
    """BTHD-view inputs (the model's natural post-RoPE layout) must
    produce identical results and grads to the contiguous path — with
    no transpose-contiguous copies."""
    torch.manual_seed(0)
    B, Hq, Hkv, T, D = 2, 8, 2, 512, 128
    base_q = torch.randn(B, T, Hq, D, device="cuda", dtype=torch.bfloat16)
    base_k = torch.randn(B, T, Hkv, D, device="cuda", dtype=torch.bfloat16)
    base_v = torch.randn(B, T, Hkv, D, device="cuda", dtype=torch.bfloat16)
    g = torch.randn(B, Hq, T, D, device="cuda", dtype=torch.bfloat16)

    def run(make):
        q = make(base_q).requires_grad_()
        k = make(base_k).requires_grad_()
        v = make(base_v).requires_grad_()
        out = ops.flash_attention(q, k, v, causal=True)
        out.backward(g)
        return out.detach().float(), q.grad.float(), k.grad.float(), v.grad.float()

    o1, dq1, dk1, dv1 = run(lambda t: t.detach().clone().transpose(1, 2))
    o2, dq2, dk2, dv2 = run(
        lambda t: t.detach().clone().transpose(1, 2).contiguous()
    )
    for a, b in ((o1, o2), (dq1, dq2), (dk1, dk2), (dv1, dv2)):
        assert torch.equal(a, b) or (a - b).abs().max().item() < 1e-6


@pytest.mark.gpu
def test_chunked_lm_head_ce_matches_plain():
    """Chunked fused lm_head+CE (never materializes full logits) must
    match the plain logits->cross_entropy path in loss and grads."""
    torch.manual_seed(0)
    N, H, V = 1024, 256, 1000
    x0 = torch.randn(N, H, device="cuda", dtype=torch.bfloat16)
    w0 = torch.randn(V, H, device="cuda", dtype=torch.bfloat16) * 0.02
    t = torch.randint(0, V, (N,), device="cuda")
    t[::7] = -100  # ignored rows

    x1 = x0.clone().requires_grad_()
    w1 = w0.clone().requires_grad_()
    loss1 = ops.lm_head_cross_entropy(x1, w1, t, chunk_rows=300)
    loss1.backward()

    x2 = x0.clone().requires_grad_()
    w2 = w0.clone().requires_grad_()
    loss2 = ops.cross_entropy(x2 @ w2.t(), t)
    loss2.backward()

    assert abs(loss1.item() - loss2.item()) < 2e-3
    def relerr(a, b):
        return ((a.float() - b.float()).norm() / (b.float().norm() + 1e-6)).item()
    assert relerr(x1.grad, x2.grad) < 0.03
    assert relerr(w1.grad, w2.grad) < 0.03


@GPU
def test_rope_strided_fused_qkv_slice():
    """The rope kernel accepts a no-copy slice of a fused-QKV GEMM
    output (arbitrary (b,t)-row stride, contiguous [H,D] tail) and
    matches the contiguous result."""
    dev = _cuda()
    torch.manual_seed(1)
    B, T, Hq, Hkv, D = 2, 16, 4, 2, 64
    cosT, sinT = ops.rope_tables(T, D, device=dev)
    fused = torch.randn(B, T, (Hq + 2 * Hkv) * D, device=dev,
                        dtype=torch.bfloat16)
    q = fused[..., : Hq * D].view(B, T, Hq, D)           # strided slice
    k = fused[..., Hq * D: (Hq + Hkv) * D].view(B, T, Hkv, D)
    assert not q.is_contiguous() and not k.is_contiguous()
    yq = ops.rope(q, cosT, sinT)
    yk = ops.rope(k, cosT, sinT)
    assert yq.is_contiguous() and yk.is_contiguous()
    assert torch.allclose(yq.float(),
                          ops.rope(q.contiguous(), cosT, sinT).float())
    assert torch.allclose(yk.float(),
                          ops.rope_ref(k.contiguous(), cosT, sinT).float(),
                          atol=2e-2, rtol=2e-2)


@GPU
def test_gemv_matches_linear():
    """Decode GEMV kernel vs F.linear across the decode shapes."""
    dev = _cuda()
    torch.manual_seed(2)
    for B, N, K in ((1, 4096, 4096), (4, 6144, 4096), (8, 4096, 14336),
                    (1, 128256, 4096)):
        x = torch.randn(B, K, device=dev, dtype=torch.bfloat16)
        w = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.02
        y = ops.linear_sb(x, w)
        yr = torch.nn.functional.linear(x.float(), w.float())
        assert torch.allclose(y.float(), yr, atol=0.3, rtol=3e-2), (B, N, K)
