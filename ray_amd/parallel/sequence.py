"""Sequence parallelism: ring attention + Ulysses all-to-all.

New capability relative to the reference (absent
there). Ring attention: each rank holds a contiguous sequence shard of
Q/K/V; K/V blocks rotate around the ring while each rank accumulates
blockwise attention with online logsumexp merging — the p2p pass
overlaps naturally with block compute on xGMI (one neighbor link).

Math is fp32 blockwise attention with explicit LSE so the merge is
exact; the fused CDNA4 flash kernel with LSE output is the planned
fast path.
"""
from __future__ import annotations

import torch
import torch.distributed as dist


def _block_attn(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                causal_mask: str) -> tuple:
    """Blockwise attention returning (out, lse).

    q [B,H,Tq,D], k/v [B,H,Tk,D]. causal_mask: "full" (attend all),
    "causal" (Tq==Tk lower-triangular), "none" (skip — caller handles).
    On GPU with head_dim 128 this runs the fused CDNA4 flash kernel
    (LSE output was built for exactly this merge)."""
    needs_grad = torch.is_grad_enabled() and (
        q.requires_grad or k.requires_grad or v.requires_grad
    )
    if (
        q.is_cuda
        and q.shape[-1] == 128
        and q.dtype == torch.bfloat16
        and not needs_grad  # lse-merging bwd lands next round
    ):
        from ray_amd import ops

        out, lse = ops.flash_attention(
            q, k, v, causal=(causal_mask == "causal"), return_lse=True
        )
        return out.float(), lse.unsqueeze(-1)
    scale = q.shape[-1] ** -0.5
    s = torch.matmul(q.float(), k.float().transpose(-1, -2)) * scale
    if causal_mask == "causal":
        Tq, Tk = s.shape[-2], s.shape[-1]
        mask = torch.ones(Tq, Tk, dtype=torch.bool, device=s.device).tril_()
        s = s.masked_fill(~mask, float("-inf"))
    lse = torch.logsumexp(s, dim=-1, keepdim=True)  # [B,H,Tq,1]
    p = torch.exp(s - lse)
    out = torch.matmul(p, v.float())
    return out, lse


def _merge(out_a, lse_a, out_b, lse_b):
    """Merge two partial attention results with logsumexp weights."""
    lse = torch.logaddexp(lse_a, lse_b)
    out = out_a * torch.exp(lse_a - lse) + out_b * torch.exp(lse_b - lse)
    return out, lse


def ring_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                   group=None, causal: bool = True) -> torch.Tensor:
    """q/k/v: this rank's sequence shard [B, H, T_local, D]; the global
    sequence is the rank-order concatenation of shards. Returns this
    rank's output shard.

    group: a torch.distributed ProcessGroup (or a
    ray_amd.util.collective Group's .pg). None -> default group.
    """
    if group is not None and hasattr(group, "pg"):
        group = group.pg
    if group is None:
        rank = dist.get_rank()
        world = dist.get_world_size()
    else:
        rank = group.rank()
        world = group.size()
    if world == 1:
        mode = "causal" if causal else "full"
        out, _ = _block_attn(q, k, v, mode)
        return out.to(q.dtype)

    cur_k, cur_v = k.contiguous(), v.contiguous()
    out = None
    lse = None
    send_to = (rank + 1) % world
    recv_from = (rank - 1) % world
    for step in range(world):
        src_block = (rank - step) % world  # whose K/V we hold now
        # exchange first? compute then pass: pipeline both
        if step < world - 1:
            nxt_k = torch.empty_like(cur_k)
            nxt_v = torch.empty_like(cur_v)
            reqs = []
            if group is None:
                reqs.append(dist.isend(cur_k, send_to))
                reqs.append(dist.isend(cur_v, send_to))
                reqs.append(dist.irecv(nxt_k, recv_from))
                reqs.append(dist.irecv(nxt_v, recv_from))
            else:
                reqs.append(group.send([cur_k], send_to, 0))
                reqs.append(group.send([cur_v], send_to, 1))
                reqs.append(group.recv([nxt_k], recv_from, 0))
                reqs.append(group.recv([nxt_v], recv_from, 1))
        if not causal:
            mode = "full"
        elif src_block == rank:
            mode = "causal"
        elif src_block < rank:
            mode = "full"
        else:
            mode = "none"  # future block: masked out entirely
        if mode != "none":
            o, l = _block_attn(q, cur_k, cur_v, mode)
            if out is None:
                out, lse = o, l
            else:
                out, lse = _merge(out, lse, o, l)
        if step < world - 1:
            for r in reqs:
                r.wait()
            cur_k, cur_v = nxt_k, nxt_v
    return out.to(q.dtype)


def ulysses_all_to_all(x: torch.Tensor, group=None,
                       scatter_dim: int = 2, gather_dim: int = 1
                       ) -> torch.Tensor:
    """DeepSpeed-Ulysses style head<->sequence all-to-all: input
    [B, H, T_local, D] sharded over sequence -> output
    [B, H/P, T_global, D] sharded over heads (or back, by swapping
    dims). xGMI's full point-to-point connectivity makes all-to-all
    cheap."""
    if group is not None and hasattr(group, "pg"):
        group = group.pg
    world = dist.get_world_size() if group is None else group.size()
    if world == 1:
        return x
    in_parts = [c.contiguous() for c in x.chunk(world, dim=gather_dim)]
    out_parts = [torch.empty_like(in_parts[0]) for _ in range(world)]
    if group is None:
        dist.all_to_all(out_parts, in_parts)
    else:
        group.alltoall(out_parts, in_parts).wait()
    return torch.cat(out_parts, dim=scatter_dim)
