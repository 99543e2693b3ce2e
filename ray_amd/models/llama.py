"""Llama-3 architecture, MI355X-first.

Hand-written HIP kernels (ray_amd.ops) for RMSNorm / RoPE / SwiGLU /
fused cross-entropy; GQA training attention on the hand-written CDNA4
flash kernels (SDPA fallback for decode/odd shapes; formerly
ROCm); plain GEMMs via nn.Linear (hipBLASLt). bf16 weights, fp32
optimizer states via ray_amd.ops.FusedAdamW.

Config llama3-8b matches Meta-Llama-3-8B: hidden 4096, 32 layers,
32 heads / 8 KV heads, ffn 14336, vocab 128256, rope base 500000.
"""
from __future__ import annotations

from dataclasses import dataclass

import os

import torch
import torch.nn as nn
import torch.nn.functional as F

from ray_amd import ops


@dataclass
class LlamaConfig:
    vocab_size: int = 128256
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_layers: int = 32
    num_heads: int = 32
    num_kv_heads: int = 8
    max_seq_len: int = 8192
    rope_base: float = 500000.0
    rms_eps: float = 1e-5
    tie_embeddings: bool = False


CONFIGS = {
    "llama3-8b": LlamaConfig(),
    "llama3-1b": LlamaConfig(
        hidden_size=2048, intermediate_size=8192, num_layers=16,
        num_heads=32, num_kv_heads=8
    ),
    # tiny config for CPU tests / smoke
    "llama-tiny": LlamaConfig(
        vocab_size=512, hidden_size=256, intermediate_size=512,
        num_layers=2, num_heads=4, num_kv_heads=2, max_seq_len=512,
    ),
}


class Attention(nn.Module):
    def __init__(self, cfg: LlamaConfig, dtype=torch.bfloat16):
        super().__init__()
        self.cfg = cfg
        self.head_dim = cfg.hidden_size // cfg.num_heads
        self.n_rep = cfg.num_heads // cfg.num_kv_heads
        # Fused QKV projection (one GEMM instead of three: x streams
        # from HBM once, and hipBLASLt's dgrad/wgrad run ~10% faster at
        # the fatter N — measured in tools/gemm_fuse_ab.py). The q/k/v
        # splits feed RoPE without a copy (strided-input rope kernel).
        self._fused_qkv = not os.environ.get("RAY_AMD_NO_FUSED_QKV")
        qd = cfg.num_heads * self.head_dim
        kvd = cfg.num_kv_heads * self.head_dim
        if self._fused_qkv:
            self.qkv_proj = nn.Linear(
                cfg.hidden_size, qd + 2 * kvd, bias=False, dtype=dtype
            )
        else:
            self.q_proj = nn.Linear(
                cfg.hidden_size, qd, bias=False, dtype=dtype)
            self.k_proj = nn.Linear(
                cfg.hidden_size, kvd, bias=False, dtype=dtype)
            self.v_proj = nn.Linear(
                cfg.hidden_size, kvd, bias=False, dtype=dtype)
        self.o_proj = nn.Linear(
            cfg.num_heads * self.head_dim, cfg.hidden_size, bias=False, dtype=dtype
        )

    def qkv(self, x, lin=None):
        """[B,T,H] -> q [B,T,Hq,D], k/v [B,T,Hkv,D] (views when fused).
        `lin(x, weight)` overrides the matmul (decode paths pass the
        GEMV fast path)."""
        B, T, _ = x.shape
        cfg = self.cfg
        hd = self.head_dim
        if self._fused_qkv:
            qkv = (self.qkv_proj(x) if lin is None
                   else lin(x, self.qkv_proj.weight))
            q, k, v = qkv.split(
                [cfg.num_heads * hd, cfg.num_kv_heads * hd,
                 cfg.num_kv_heads * hd], dim=-1)
            return (q.view(B, T, cfg.num_heads, hd),
                    k.view(B, T, cfg.num_kv_heads, hd),
                    v.view(B, T, cfg.num_kv_heads, hd))
        if lin is None:
            return (self.q_proj(x).view(B, T, cfg.num_heads, hd),
                    self.k_proj(x).view(B, T, cfg.num_kv_heads, hd),
                    self.v_proj(x).view(B, T, cfg.num_kv_heads, hd))
        return (lin(x, self.q_proj.weight).view(B, T, cfg.num_heads, hd),
                lin(x, self.k_proj.weight).view(B, T, cfg.num_kv_heads, hd),
                lin(x, self.v_proj.weight).view(B, T, cfg.num_kv_heads, hd))

    def forward(self, x, cosT, sinT, kv_cache=None, pos0: int = 0):
        B, T, H = x.shape
        cfg = self.cfg
        q, k, v = self.qkv(x)
        if self._fused_qkv:
            v = v.contiguous()  # k/q gather happens inside rope
        q = ops.rope(q, cosT[pos0 : pos0 + T], sinT[pos0 : pos0 + T])
        k = ops.rope(k, cosT[pos0 : pos0 + T], sinT[pos0 : pos0 + T])
        q = q.transpose(1, 2)  # [B, Hq, T, D]
        k = k.transpose(1, 2)
        v = v.transpose(1, 2)
        if kv_cache is not None:
            k, v = kv_cache.update(k, v, pos0)
        is_causal = kv_cache is None or T > 1
        # training self-attention runs the hand-written CDNA4 flash
        # kernels (fwd v6 + fa_bwd v4: measured at parity with AOTriton
        # round-trip, profiles/flash_attn_v6_bench.log); decode/cache
        # shapes keep SDPA
        if (
            kv_cache is None
            and q.is_cuda
            and q.dtype == torch.bfloat16
            and self.head_dim == 128
            and T % 128 == 0
            and not os.environ.get("RAY_AMD_ATTN_SDPA")
        ):
            out = ops.flash_attention(q, k, v, causal=True)
        else:
            out = F.scaled_dot_product_attention(
                q, k, v, is_causal=is_causal, enable_gqa=True
            )
        out = out.transpose(1, 2).reshape(B, T, -1)
        return self.o_proj(out)


class MLP(nn.Module):
    def __init__(self, cfg: LlamaConfig, dtype=torch.bfloat16):
        super().__init__()
        self.gate_proj = nn.Linear(
            cfg.hidden_size, cfg.intermediate_size, bias=False, dtype=dtype
        )
        self.up_proj = nn.Linear(
            cfg.hidden_size, cfg.intermediate_size, bias=False, dtype=dtype
        )
        self.down_proj = nn.Linear(
            cfg.intermediate_size, cfg.hidden_size, bias=False, dtype=dtype
        )

    def forward(self, x):
        return self.down_proj(ops.swiglu(self.gate_proj(x), self.up_proj(x)))


class Block(nn.Module):
    def __init__(self, cfg: LlamaConfig, dtype=torch.bfloat16):
        super().__init__()
        self.attn_norm = ops.RMSNorm(cfg.hidden_size, cfg.rms_eps, dtype)
        self.attn = Attention(cfg, dtype)
        self.mlp_norm = ops.RMSNorm(cfg.hidden_size, cfg.rms_eps, dtype)
        self.mlp = MLP(cfg, dtype)

    def forward(self, x, cosT, sinT, kv_cache=None, pos0=0):
        x = x + self.attn(self.attn_norm(x), cosT, sinT, kv_cache, pos0)
        x = x + self.mlp(self.mlp_norm(x))
        return x


class LlamaModel(nn.Module):
    def __init__(self, cfg: LlamaConfig, dtype=torch.bfloat16,
                 gradient_checkpointing: bool = False):
        super().__init__()
        self.cfg = cfg
        self.dtype = dtype
        self.embed = nn.Embedding(cfg.vocab_size, cfg.hidden_size, dtype=dtype)
        self.layers = nn.ModuleList(Block(cfg, dtype) for _ in range(cfg.num_layers))
        self.final_norm = ops.RMSNorm(cfg.hidden_size, cfg.rms_eps, dtype)
        self.lm_head = nn.Linear(
            cfg.hidden_size, cfg.vocab_size, bias=False, dtype=dtype
        )
        if cfg.tie_embeddings:
            self.lm_head.weight = self.embed.weight
        cosT, sinT = ops.rope_tables(
            cfg.max_seq_len, cfg.hidden_size // cfg.num_heads, cfg.rope_base
        )
        self.register_buffer("cosT", cosT, persistent=False)
        self.register_buffer("sinT", sinT, persistent=False)
        self.gradient_checkpointing = gradient_checkpointing
        self.apply(self._init_weights)

    def _init_weights(self, m):
        if isinstance(m, nn.Linear):
            nn.init.normal_(m.weight, std=0.02)
        elif isinstance(m, nn.Embedding):
            nn.init.normal_(m.weight, std=0.02)

    def forward(self, tokens, targets=None, kv_caches=None, pos0: int = 0):
        x = self.embed(tokens)
        for i, layer in enumerate(self.layers):
            cache = kv_caches[i] if kv_caches is not None else None
            if self.gradient_checkpointing and self.training:
                x = torch.utils.checkpoint.checkpoint(
                    layer, x, self.cosT, self.sinT, use_reentrant=False
                )
            else:
                x = layer(x, self.cosT, self.sinT, cache, pos0)
        x = self.final_norm(x)
        if targets is not None:
            if x.is_cuda and os.environ.get("RAY_AMD_CHUNKED_CE"):
                # opt-in chunked fused lm_head+CE: saves ~15 GB peak
                # (the full logits tensor is never materialized) at the
                # cost of one extra lm_head GEMM (~2% step) — for
                # memory-bound configs; measured slower at mb8
                return ops.lm_head_cross_entropy(
                    x.reshape(-1, self.cfg.hidden_size),
                    self.lm_head.weight,
                    targets.reshape(-1),
                )
            logits = self.lm_head(x)
            return ops.cross_entropy(
                logits.view(-1, self.cfg.vocab_size), targets.reshape(-1)
            )
        return self.lm_head(x)

    def num_params(self) -> int:
        return sum(p.numel() for p in self.parameters())


class GraphedDecoder:
    """hipGraph-captured single-token decode (north-star config 5:
    "hipGraph-captured decode"; HIP graphs instead of a tracing
    compiler).

    The decode step is made shape- and address-static so one capture
    replays for every token: KV caches are preallocated at max_T, the
    new K/V row is written via index_copy_ with a device position
    tensor, and attention runs over the full cache with an additive
    mask buffer updated in place. All tensors the graph reads are
    static buffers; per-token updates are in-place copies."""

    def __init__(self, model: "LlamaModel", batch_size: int, max_T: int,
                 device):
        self.m = model
        cfg = model.cfg
        self.B = batch_size
        self.max_T = max_T
        self.device = device
        hd = cfg.hidden_size // cfg.num_heads
        dt = model.dtype
        self.cache_k = [
            torch.zeros(batch_size, cfg.num_kv_heads, max_T, hd,
                        device=device, dtype=dt)
            for _ in range(cfg.num_layers)
        ]
        self.cache_v = [torch.zeros_like(k) for k in self.cache_k]
        # static IO buffers
        self.in_tok = torch.zeros(batch_size, 1, dtype=torch.long, device=device)
        self.pos = torch.zeros(1, dtype=torch.long, device=device)
        self.mask = torch.full((1, 1, 1, max_T), float("-inf"),
                               device=device, dtype=torch.float32)
        self.out_logits = None
        self.graph = None

    def _step(self):
        m = self.m
        cfg = m.cfg
        B = self.B
        hd = cfg.hidden_size // cfg.num_heads
        x = m.embed(self.in_tok)  # [B,1,H]
        cos = m.cosT.index_select(0, self.pos).view(1, 1, 1, hd // 2)
        sin = m.sinT.index_select(0, self.pos).view(1, 1, 1, hd // 2)
        lin = ops.linear_sb  # GEMV fast path (hipBLASLt is 4x off at M<=8)
        for li, layer in enumerate(m.layers):
            h = layer.attn_norm(x)
            at = layer.attn
            q, k, v = at.qkv(h, lin=lin)
            q = _rope_one(q, cos, sin)
            k = _rope_one(k, cos, sin)
            # static-address cache update
            self.cache_k[li].index_copy_(
                2, self.pos, k.permute(0, 2, 1, 3).to(self.cache_k[li].dtype)
            )
            self.cache_v[li].index_copy_(
                2, self.pos, v.permute(0, 2, 1, 3).to(self.cache_v[li].dtype)
            )
            attn = F.scaled_dot_product_attention(
                q.permute(0, 2, 1, 3),
                self.cache_k[li],
                self.cache_v[li],
                attn_mask=self.mask.to(q.dtype),
                enable_gqa=True,
            )
            x = x + lin(attn.permute(0, 2, 1, 3).reshape(B, 1, -1),
                        at.o_proj.weight)
            hm = layer.mlp_norm(x)
            mp = layer.mlp
            x = x + lin(ops.swiglu(lin(hm, mp.gate_proj.weight),
                                   lin(hm, mp.up_proj.weight)),
                        mp.down_proj.weight)
        x = m.final_norm(x)
        return lin(x, m.lm_head.weight)[:, 0]

    @torch.no_grad()  # grad mode gates the GEMV fast path (linear_sb)
    def capture(self):
        assert self.device.type == "cuda"
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):  # warmup allocations
                self.out_logits = self._step()
        torch.cuda.current_stream().wait_stream(s)
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.out_logits = self._step()

    @torch.no_grad()
    def decode(self, tok: torch.Tensor, position: int) -> torch.Tensor:
        self.in_tok.copy_(tok.view(self.B, 1))
        self.pos.fill_(position)
        self.mask[..., : position + 1] = 0.0
        if self.graph is not None:
            self.graph.replay()
            return self.out_logits
        return self._step()


def _rope_one(x, cos, sin):
    """RoPE for a single position; x [B,1,Hn,D], cos/sin [1,1,1,D/2]."""
    half = x.shape[-1] // 2
    x1 = x[..., :half].float()
    x2 = x[..., half:].float()
    c = cos.to(torch.float32)
    s = sin.to(torch.float32)
    return torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], dim=-1).to(x.dtype)


class BatchedDecoder:
    """Single-token decode for a BATCH of independent sequences at
    per-slot positions — the substrate for continuous batching
    (reference: serve LLM's vLLM engine loop). Same static-buffer
    design as GraphedDecoder but pos/mask are per-row; eager execution
    (hipGraph capture of the batched step lands next round)."""

    def __init__(self, model: "LlamaModel", batch_size: int, max_T: int,
                 device):
        self.m = model
        cfg = model.cfg
        self.B = batch_size
        self.max_T = max_T
        self.device = device
        hd = cfg.hidden_size // cfg.num_heads
        dt = model.dtype
        self.cache_k = [
            torch.zeros(batch_size, cfg.num_kv_heads, max_T, hd,
                        device=device, dtype=dt)
            for _ in range(cfg.num_layers)
        ]
        self.cache_v = [torch.zeros_like(k) for k in self.cache_k]
        self.mask = torch.full((batch_size, 1, 1, max_T), float("-inf"),
                               device=device, dtype=torch.float32)
        self._rows = torch.arange(batch_size, device=device)

    def set_slot_len(self, slot: int, length: int):
        self.mask[slot, ..., :length] = 0.0
        self.mask[slot, ..., length:] = float("-inf")

    def prefill_slot(self, slot: int, toks: torch.Tensor):
        """Run the prompt through the model, filling this slot's cache
        rows; returns last-position logits [V]."""
        caches = [
            _SlotCacheView(self.cache_k[li], self.cache_v[li], slot)
            for li in range(self.m.cfg.num_layers)
        ]
        logits = self.m(toks.view(1, -1), kv_caches=caches, pos0=0)[0, -1]
        self.set_slot_len(slot, toks.numel())
        return logits

    @torch.no_grad()
    def decode(self, toks: torch.Tensor, pos: torch.Tensor):
        """toks [B] last tokens, pos [B] their positions; returns
        logits [B, V] (inactive slots produce garbage rows)."""
        m = self.m
        cfg = m.cfg
        B = self.B
        hd = cfg.hidden_size // cfg.num_heads
        x = m.embed(toks.view(B, 1))
        cos = m.cosT.index_select(0, pos).view(B, 1, 1, hd // 2)
        sin = m.sinT.index_select(0, pos).view(B, 1, 1, hd // 2)
        lin = ops.linear_sb  # GEMV fast path for small decode batches
        for li, layer in enumerate(m.layers):
            h = layer.attn_norm(x)
            at = layer.attn
            q, k, v = at.qkv(h, lin=lin)
            q = _rope_one(q, cos, sin)
            k = _rope_one(k, cos, sin)
            # per-row cache write at each slot's own position
            self.cache_k[li][self._rows, :, pos] = (
                k[:, 0].to(self.cache_k[li].dtype)
            )
            self.cache_v[li][self._rows, :, pos] = (
                v[:, 0].to(self.cache_v[li].dtype)
            )
            attn = F.scaled_dot_product_attention(
                q.permute(0, 2, 1, 3),
                self.cache_k[li],
                self.cache_v[li],
                attn_mask=self.mask.to(q.dtype),
                enable_gqa=True,
            )
            x = x + lin(attn.permute(0, 2, 1, 3).reshape(B, 1, -1),
                        at.o_proj.weight)
            hm = layer.mlp_norm(x)
            mp = layer.mlp
            x = x + lin(ops.swiglu(lin(hm, mp.gate_proj.weight),
                                   lin(hm, mp.up_proj.weight)),
                        mp.down_proj.weight)
        x = m.final_norm(x)
        return lin(x, m.lm_head.weight)[:, 0]


class _SlotCacheView:
    """KV-cache adapter targeting one slot of a BatchedDecoder."""

    def __init__(self, k_buf, v_buf, slot: int):
        self.k_buf = k_buf
        self.v_buf = v_buf
        self.slot = slot
        self.len = 0

    def update(self, k, v, pos0):
        T = k.size(2)
        self.k_buf[self.slot : self.slot + 1, :, pos0 : pos0 + T] = k.to(
            self.k_buf.dtype
        )
        self.v_buf[self.slot : self.slot + 1, :, pos0 : pos0 + T] = v.to(
            self.v_buf.dtype
        )
        self.len = pos0 + T
        return (
            self.k_buf[self.slot : self.slot + 1, :, : self.len],
            self.v_buf[self.slot : self.slot + 1, :, : self.len],
        )


class KVCache:
    """Per-layer KV cache for decode, preallocated in HBM."""

    def __init__(self, B, max_T, n_kv, head_dim, device, dtype=torch.bfloat16):
        self.k = torch.zeros(B, n_kv, max_T, head_dim, device=device, dtype=dtype)
        self.v = torch.zeros(B, n_kv, max_T, head_dim, device=device, dtype=dtype)
        self.len = 0

    def update(self, k, v, pos0):
        T = k.size(2)
        self.k[:, :, pos0 : pos0 + T] = k
        self.v[:, :, pos0 : pos0 + T] = v
        self.len = pos0 + T
        return self.k[:, :, : self.len], self.v[:, :, : self.len]
