"""ray_amd.ops — hand-written CDNA4 HIP kernels with torch autograd.

Each op has a plain-PyTorch fp32 reference (`*_ref`) used on CPU and in
numerics tests. On a GPU box the HIP extension is REQUIRED: if a CUDA
tensor reaches an op and `ray_amd._hip_ops` is missing, we raise rather
than silently falling back to eager torch.
"""
from __future__ import annotations

import torch

try:
    from ray_amd import _hip_ops as _K

    HAVE_HIP_OPS = True
except ImportError:  # CPU-only environment without a built extension
    _K = None
    HAVE_HIP_OPS = False


def _require_ext(name: str):
    if _K is None:
        raise RuntimeError(
            f"ray_amd HIP extension not built but {name} was called on a GPU "
            "tensor. Run `python ray_amd/csrc/build.py`."
        )


# --------------------------------------------------------------------------
# RMSNorm
# --------------------------------------------------------------------------


def rmsnorm_ref(x: torch.Tensor, w: torch.Tensor, eps: float = 1e-5):
    xf = x.float()
    inv = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (xf * inv * w.float()).to(x.dtype)


class _RMSNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, eps):
        _require_ext("rmsnorm")
        y, inv = _K.rmsnorm_fwd(x.contiguous(), w.contiguous(), eps)
        ctx.save_for_backward(x, w, inv)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, inv = ctx.saved_tensors
        dx, dw = _K.rmsnorm_bwd(dy.contiguous(), x, w, inv)
        return dx, dw.to(w.dtype), None


def rmsnorm(x: torch.Tensor, w: torch.Tensor, eps: float = 1e-5):
    if not x.is_cuda:
        # autograd-capable reference path
        xf = x.float()
        inv = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
        return (xf * inv * w.float()).to(x.dtype)
    return _RMSNorm.apply(x, w, eps)


class RMSNorm(torch.nn.Module):
    def __init__(self, dim: int, eps: float = 1e-5, dtype=torch.bfloat16):
        super().__init__()
        self.weight = torch.nn.Parameter(torch.ones(dim, dtype=dtype))
        self.eps = eps

    def forward(self, x):
        return rmsnorm(x, self.weight, self.eps)


# --------------------------------------------------------------------------
# SwiGLU
# --------------------------------------------------------------------------


def swiglu_ref(a: torch.Tensor, b: torch.Tensor):
    return (torch.nn.functional.silu(a.float()) * b.float()).to(a.dtype)


class _SwiGLU(torch.autograd.Function):
    @staticmethod
    def forward(ctx, a, b):
        _require_ext("swiglu")
        a = a.contiguous()
        b = b.contiguous()
        ctx.save_for_backward(a, b)
        return _K.swiglu_fwd(a, b)

    @staticmethod
    def backward(ctx, dy):
        a, b = ctx.saved_tensors
        da, db = _K.swiglu_bwd(dy.contiguous(), a, b)
        return da, db


def swiglu(a: torch.Tensor, b: torch.Tensor):
    if not a.is_cuda:
        return torch.nn.functional.silu(a) * b
    return _SwiGLU.apply(a, b)


# --------------------------------------------------------------------------
# RoPE
# --------------------------------------------------------------------------


def rope_tables(T: int, D: int, base: float = 500000.0, device="cpu"):
    """Host-precomputed cos/sin [T, D/2] fp32 (guide §B: no device trig)."""
    half = D // 2
    inv_freq = 1.0 / (
        base ** (torch.arange(0, half, dtype=torch.float64) / half)
    )
    t = torch.arange(T, dtype=torch.float64)
    ang = torch.outer(t, inv_freq)
    return (
        ang.cos().float().contiguous().to(device),
        ang.sin().float().contiguous().to(device),
    )


def rope_ref(x: torch.Tensor, cosT: torch.Tensor, sinT: torch.Tensor):
    """x: [B, T, Hn, D]; rotate halves (LLaMA convention)."""
    B, T, Hn, D = x.shape
    half = D // 2
    x1 = x[..., :half].float()
    x2 = x[..., half:].float()
    c = cosT[:T].view(1, T, 1, half)
    s = sinT[:T].view(1, T, 1, half)
    return torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], dim=-1).to(x.dtype)


class _Rope(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, cosT, sinT, n_heads, T):
        _require_ext("rope")
        ctx.meta = (cosT, sinT, n_heads, T)
        # fused-QKV slices ([H,D]-contiguous, strided (b,t) rows) go
        # straight to the kernel — RoPE doubles as the gather
        if not (x.dim() == 4 and x.stride(3) == 1
                and x.stride(2) == x.size(3)
                and x.stride(0) == x.size(1) * x.stride(1)):
            x = x.contiguous()
        return _K.rope_apply(x, cosT, sinT, n_heads, T, 1)

    @staticmethod
    def backward(ctx, dy):
        cosT, sinT, n_heads, T = ctx.meta
        dx = _K.rope_apply(dy.contiguous(), cosT, sinT, n_heads, T, -1)
        return dx, None, None, None, None


def rope(x: torch.Tensor, cosT: torch.Tensor, sinT: torch.Tensor):
    """x: [B, T, Hn, D] bf16."""
    B, T, Hn, D = x.shape
    if not x.is_cuda:
        return rope_ref(x, cosT, sinT)
    return _Rope.apply(x, cosT[:T].contiguous(), sinT[:T].contiguous(), Hn, T)


def linear_sb(x: torch.Tensor, weight: torch.Tensor) -> torch.Tensor:
    """Small-batch linear for decode (csrc/hip/gemv.hip): one wave per
    output row streams W once. Measured vs hipBLASLt
    (tools/gemv_ab.py): wins 1.4-2x at rows<=2 with N<=8192 (the
    qkv/o projections at decode batch 1-2); hipBLASLt already runs
    near-bandwidth (5.7+ TB/s) at larger N/K, so those shapes fall
    back. VALU-bound above rows=2 (each W element costs `rows` fmas)."""
    K = x.shape[-1]
    rows = x.numel() // K
    if (
        _K is not None
        and x.is_cuda
        and x.dtype == torch.bfloat16
        and weight.dtype == torch.bfloat16
        and rows <= 2
        and weight.shape[0] <= 8192
        and K % 512 == 0
        and not torch.is_grad_enabled()
    ):
        y = _K.gemv(x.reshape(rows, K).contiguous(), weight)
        return y.view(*x.shape[:-1], weight.shape[0])
    return torch.nn.functional.linear(x, weight)


# --------------------------------------------------------------------------
# Cross entropy (fused, bf16 logits)
# --------------------------------------------------------------------------


def cross_entropy_ref(logits: torch.Tensor, target: torch.Tensor):
    return torch.nn.functional.cross_entropy(
        logits.float(), target.long(), ignore_index=-100, reduction="mean"
    )


class _FusedCE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, target):
        _require_ext("cross_entropy")
        logits = logits.contiguous()
        t32 = target.to(torch.int32).contiguous()
        loss, mx, lse = _K.ce_fwd(logits, t32)
        nvalid = (t32 >= 0).sum().clamp(min=1)
        ctx.save_for_backward(logits, t32, mx, lse, nvalid)
        return loss.sum() / nvalid.float()

    @staticmethod
    def backward(ctx, dloss):
        logits, t32, mx, lse, nvalid = ctx.saved_tensors
        per_row = (dloss / nvalid.float()).expand(logits.size(0)).contiguous()
        dl = _K.ce_bwd(logits, t32, mx, lse, per_row)
        return dl, None


def cross_entropy(logits: torch.Tensor, target: torch.Tensor):
    """Mean CE over rows where target != -100. logits [N, V] bf16."""
    if not logits.is_cuda:
        return cross_entropy_ref(logits, target)
    return _FusedCE.apply(logits, target)


class _ChunkedLmHeadCE(torch.autograd.Function):
    """Fused lm_head + cross entropy, row-chunked: the [N, V] logits
    are never materialized whole (N=B*T=32k, V=128k bf16 = 8.4 GB at
    the flagship shape — plus its stored-activation copy). Forward
    computes the loss per chunk; backward RECOMPUTES each chunk's
    logits and uses the fused CE kernels for dlogits, folding straight
    into dX and a fp32 dW accumulator. Costs one extra lm_head GEMM,
    saves ~15 GB peak (headroom for larger micro-batches)."""

    @staticmethod
    def forward(ctx, x, weight, target, chunk_rows):
        _require_ext("cross_entropy")
        x = x.contiguous()
        t32 = target.to(torch.int32).contiguous()
        N = x.shape[0]
        nvalid = (t32 >= 0).sum().clamp(min=1)
        loss_sum = torch.zeros((), dtype=torch.float32, device=x.device)
        for s in range(0, N, chunk_rows):
            e = min(s + chunk_rows, N)
            logits = (x[s:e] @ weight.t()).contiguous()
            l, mx, lse = _K.ce_fwd(logits, t32[s:e])
            loss_sum += l.sum()
        ctx.save_for_backward(x, weight, t32, nvalid)
        ctx.chunk = chunk_rows
        return loss_sum / nvalid.float()

    @staticmethod
    def backward(ctx, dloss):
        x, w, t32, nvalid = ctx.saved_tensors
        N = x.shape[0]
        chunk = ctx.chunk
        dx = torch.empty_like(x)
        dw_acc = torch.zeros_like(w, dtype=torch.float32)
        scale = dloss / nvalid.float()
        for s in range(0, N, chunk):
            e = min(s + chunk, N)
            logits = (x[s:e] @ w.t()).contiguous()
            _, mx, lse = _K.ce_fwd(logits, t32[s:e])
            per_row = scale.expand(e - s).contiguous()
            dl = _K.ce_bwd(logits, t32[s:e], mx, lse, per_row)
            dx[s:e] = dl @ w
            dw_acc += (dl.t() @ x[s:e]).float()
        return dx, dw_acc.to(w.dtype), None, None


def lm_head_cross_entropy(x: torch.Tensor, weight: torch.Tensor,
                          target: torch.Tensor,
                          chunk_rows: int = 8192) -> torch.Tensor:
    """Chunked fused projection + CE; x [N, H] bf16, weight [V, H]."""
    if not x.is_cuda:
        return cross_entropy_ref(x.float() @ weight.t().float(), target)
    return _ChunkedLmHeadCE.apply(x, weight, target, chunk_rows)


# --------------------------------------------------------------------------
# Fused AdamW
# --------------------------------------------------------------------------


class FusedAdamW:
    """AdamW with fp32 states (+ fp32 master weights for bf16 params),
    one fused HIP kernel launch per tensor. CPU fallback uses torch ops."""

    def __init__(self, params, lr=1e-4, betas=(0.9, 0.95), eps=1e-8,
                 weight_decay=0.1):
        self.params = [p for p in params if p.requires_grad]
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.wd = weight_decay
        self.t = 0
        self.state = {}
        for p in self.params:
            st = {"m": torch.zeros_like(p, dtype=torch.float32),
                  "v": torch.zeros_like(p, dtype=torch.float32)}
            if p.dtype == torch.bfloat16:
                st["master"] = p.detach().float().clone()
            self.state[id(p)] = st

    @property
    def param_groups(self):
        return [{"params": self.params, "lr": self.lr}]

    def zero_grad(self, set_to_none=True):
        for p in self.params:
            if set_to_none:
                p.grad = None
            elif p.grad is not None:
                p.grad.zero_()

    @torch.no_grad()
    def step(self, grad_scale: float = 1.0):
        self.t += 1
        for p in self.params:
            if p.grad is None:
                continue
            st = self.state[id(p)]
            if p.is_cuda:
                _require_ext("adamw")
                _K.adamw_step(
                    p.data, p.grad.contiguous(), st["m"], st["v"],
                    st.get("master", st["m"]), self.lr, self.beta1,
                    self.beta2, self.eps, self.wd, self.t, grad_scale,
                )
            else:
                g = p.grad.float() * grad_scale
                st["m"].mul_(self.beta1).add_(g, alpha=1 - self.beta1)
                st["v"].mul_(self.beta2).addcmul_(g, g, value=1 - self.beta2)
                bc1 = 1 - self.beta1 ** self.t
                bc2 = 1 - self.beta2 ** self.t
                pf = st.get("master", None)
                if pf is None:
                    pf = p.data.float()
                upd = (st["m"] / bc1) / ((st["v"] / bc2).sqrt() + self.eps)
                pf -= self.lr * (upd + self.wd * pf)
                if "master" in st:
                    p.data.copy_(pf.to(p.dtype))
                else:
                    p.data.copy_(pf)

    def state_dict(self):
        return {"t": self.t, "lr": self.lr}

    def load_state_dict(self, sd):
        self.t = sd.get("t", 0)
        self.lr = sd.get("lr", self.lr)


# --------------------------------------------------------------------------
# RL scans
# --------------------------------------------------------------------------


def gae_ref(rewards, values, cont, gamma, lam):
    """rewards/cont [T,B] fp32, values [T+1,B]. Returns (adv, vtarg)."""
    T, B = rewards.shape
    adv = torch.zeros_like(rewards)
    running = torch.zeros(B, dtype=rewards.dtype)
    for t in reversed(range(T)):
        delta = rewards[t] + gamma * cont[t] * values[t + 1] - values[t]
        running = delta + gamma * lam * cont[t] * running
        adv[t] = running
    return adv, adv + values[:-1]


def gae(rewards, values, cont, gamma=0.99, lam=0.95):
    if not rewards.is_cuda:
        return gae_ref(rewards, values, cont, gamma, lam)
    _require_ext("gae")
    out = _K.gae(rewards.contiguous(), values.contiguous(),
                 cont.contiguous(), gamma, lam)
    return out[0], out[1]


def vtrace_ref(log_rhos, rewards, values, cont, gamma, rho_clip=1.0,
               c_clip=1.0, rho_pg_clip=1.0):
    """Reference recursion (cf. vtrace_torch_v2.py:73 semantics)."""
    T, B = rewards.shape
    rhos = log_rhos.exp()
    rho = rhos.clamp(max=rho_clip)
    cs = rhos.clamp(max=c_clip)
    vs = torch.zeros_like(rewards)
    diff = torch.zeros(B, dtype=rewards.dtype)
    for t in reversed(range(T)):
        delta = rho[t] * (rewards[t] + gamma * cont[t] * values[t + 1] - values[t])
        cur = delta + gamma * cont[t] * cs[t] * diff
        vs[t] = values[t] + cur
        diff = cur
    pg = torch.zeros_like(rewards)
    for t in range(T):
        vs_next = values[t + 1] if t == T - 1 else vs[t + 1]
        pg[t] = rhos[t].clamp(max=rho_pg_clip) * (
            rewards[t] + gamma * cont[t] * vs_next - values[t]
        )
    return vs, pg


def vtrace(log_rhos, rewards, values, cont, gamma=0.99, rho_clip=1.0,
           c_clip=1.0, rho_pg_clip=1.0):
    if not rewards.is_cuda:
        return vtrace_ref(log_rhos, rewards, values, cont, gamma, rho_clip,
                          c_clip, rho_pg_clip)
    _require_ext("vtrace")
    out = _K.vtrace(log_rhos.contiguous(), rewards.contiguous(),
                    values.contiguous(), cont.contiguous(), gamma,
                    rho_clip, c_clip, rho_pg_clip)
    return out[0], out[1]


# --------------------------------------------------------------------------
# Image normalize (Data preprocessing)
# --------------------------------------------------------------------------


def img_normalize_ref(x_u8, mean, std):
    xf = x_u8.float() / 255.0
    out = (xf - mean.view(1, 1, 1, -1)) / std.view(1, 1, 1, -1)
    return out.permute(0, 3, 1, 2).contiguous().bfloat16()


def img_normalize(x_u8: torch.Tensor, mean: torch.Tensor, std: torch.Tensor):
    """uint8 NHWC -> bf16 NCHW normalized."""
    if not x_u8.is_cuda:
        return img_normalize_ref(x_u8, mean, std)
    _require_ext("img_normalize")
    return _K.img_normalize(
        x_u8.contiguous(), mean.float().contiguous(),
        (1.0 / std.float()).contiguous()
    )


# --------------------------------------------------------------------------
# Flash attention (forward; fused CDNA4 kernel with LSE output)
# --------------------------------------------------------------------------


def flash_attention_ref(q, k, v, causal=True):
    """fp32 reference; q [B,Hq,T,D], k/v [B,Hkv,Tk,D] (GQA)."""
    import math

    B, Hq, T, D = q.shape
    Hkv = k.shape[1]
    rep = Hq // Hkv
    kf = k.float().repeat_interleave(rep, dim=1)
    vf = v.float().repeat_interleave(rep, dim=1)
    s = torch.matmul(q.float(), kf.transpose(-1, -2)) / math.sqrt(D)
    if causal:
        Tk = k.shape[2]
        mask = torch.ones(T, Tk, dtype=torch.bool, device=q.device).tril_(
            Tk - T
        )
        s = s.masked_fill(~mask, float("-inf"))
    lse = torch.logsumexp(s, dim=-1)
    out = torch.matmul(torch.softmax(s, -1), vf)
    return out.to(q.dtype), lse


def _is_bthd_view(t):
    """[B,H,T,D] tensor that is a transpose view of [B,T,H,D] storage
    (the model's natural post-RoPE layout) — the kernels read it
    directly, no transpose-contiguous copy."""
    B, H, T, D = t.shape
    st = t.stride()
    return (st[3] == 1 and st[1] == D and st[2] == H * D
            and st[0] == H * T * D)


class _FlashAttnFn(torch.autograd.Function):
    """Training flash attention: v5 forward + FA2-style two-kernel
    backward (csrc/hip/fa_bwd.hip). Requires T == Tk, T % 128 == 0,
    D == 128 (the Llama training shape)."""

    @staticmethod
    def forward(ctx, q, k, v, causal):
        T = q.shape[2]
        bthd = (T % 256 == 0 and _is_bthd_view(q) and _is_bthd_view(k)
                and _is_bthd_view(v))
        if bthd:
            qc, kc, vc = q, k, v
        else:
            qc, kc, vc = q.contiguous(), k.contiguous(), v.contiguous()
        out, lse = _K.flash_attn_fwd(qc, kc, vc, causal, 0, True)
        ctx.save_for_backward(qc, kc, vc, out, lse)
        ctx.causal = causal
        ctx.bthd = bthd
        return out

    @staticmethod
    def backward(ctx, d_out):
        q, k, v, out, lse = ctx.saved_tensors
        if ctx.bthd:
            if not _is_bthd_view(d_out):
                d_out = d_out.transpose(1, 2).contiguous().transpose(1, 2)
        else:
            d_out = d_out.contiguous()
        dq, dk, dv = _K.flash_attn_bwd(
            q, k, v, out, d_out, lse, ctx.causal
        )
        return dq, dk, dv, None


def flash_attention(q, k, v, causal: bool = True, q_offset: int = 0,
                    return_lse: bool = False):
    """Fused CDNA4 flash attention. q [B,Hq,T,128] bf16 (GQA k/v
    [B,Hkv,Tk,128]). Inference pads T to 128 internally; the
    differentiable path (any input requires_grad) additionally needs
    T == Tk and T % 128 == 0 and supports backward via the HIP
    fa_bwd kernels."""
    if not q.is_cuda:
        out, lse = flash_attention_ref(q, k, v, causal)
        return (out, lse) if return_lse else out
    _require_ext("flash_attention")
    B, Hq, T, D = q.shape
    needs_grad = torch.is_grad_enabled() and (
        q.requires_grad or k.requires_grad or v.requires_grad
    )
    if needs_grad:
        if return_lse or q_offset:
            raise NotImplementedError(
                "flash_attention backward does not support return_lse/"
                "q_offset (ring-attention bwd lands next round)"
            )
        if T % 128 != 0 or T != k.shape[2]:
            raise NotImplementedError(
                "flash_attention backward requires T == Tk and "
                "T % 128 == 0"
            )
        return _FlashAttnFn.apply(q, k, v, causal)
    pad = (-T) % 128
    if pad:
        q = torch.nn.functional.pad(q, (0, 0, 0, pad))
    if not pad and T % 256 == 0 and _is_bthd_view(q) and _is_bthd_view(k) \
            and _is_bthd_view(v):
        r = _K.flash_attn_fwd(q, k, v, causal, q_offset, return_lse)
    else:
        r = _K.flash_attn_fwd(q.contiguous(), k.contiguous(),
                              v.contiguous(), causal, q_offset, return_lse)
    out = r[0][:, :, :T] if pad else r[0]
    if return_lse:
        lse = r[1][:, :, :T] if pad else r[1]
        return out, lse
    return out
