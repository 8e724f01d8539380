"""Ring attention (context parallelism) over RCCL point-to-point.

SURVEY.md §5 flags ring/CP attention as the gap past Ulysses SP (which
tops out at num_kv_heads-way sharding, ~8 on one xGMI node): each rank
holds a contiguous sequence chunk of Q/K/V, K/V blocks travel around the
ring (rank r sends to r+1, receives from r-1 — on xGMI every hop is a
dedicated point-to-point link, so the ring IS the topology-native
pattern), and softmax is accumulated blockwise with the flash-attention
running max/denominator so the result is EXACTLY full attention over the
global sequence.

This module is the primitive (forward + backward as an autograd.Function,
bit-checked against full attention in tests/test_context_parallel.py);
GroupedQueryAttention uses it when the mesh is built with
sp_mode="ring" (`--sp-mode ring`), which lifts Ulysses' head-count
divisibility limit. Fusing the block step with a CDNA4 flash kernel and
overlapping the ring hop with compute are round-2 (ROADMAP).

Layout: q, k, v are [B, H, S_local, D]; the global sequence is the rank-
order concatenation. Causality is block-causal: a K/V block from rank
j < r is fully visible to rank r's queries, j == r is locally causal,
j > r contributes nothing. GQA callers expand K/V heads first.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.distributed as dist


def _ring_pass(tensors, group) -> Tuple[torch.Tensor, ...]:
    """Send `tensors` to rank+1, receive same-shape tensors from rank-1."""
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    ranks = dist.get_process_group_ranks(group) if group is not None \
        else list(range(world))
    nxt = ranks[(rank + 1) % world]
    prv = ranks[(rank - 1) % world]
    outs = []
    works = []
    for t in tensors:
        t = t.contiguous()
        works.append(dist.isend(t, nxt, group=group))
        r = torch.empty_like(t)
        dist.recv(r, prv, group=group)
        outs.append(r)
    for w in works:
        w.wait()
    return tuple(outs)


def _block_update(q, kj, vj, acc, row_max, denom, scale, local_causal):
    """One flash-style blockwise softmax accumulation step."""
    scores = torch.matmul(q, kj.transpose(-1, -2)) * scale    # [B,H,Sq,Skv]
    if local_causal:
        Sq, Skv = scores.shape[-2], scores.shape[-1]
        causal = torch.ones(Sq, Skv, dtype=torch.bool,
                            device=scores.device).tril()
        scores = scores.masked_fill(~causal, float("-inf"))
    blk_max = scores.amax(dim=-1)
    new_max = torch.maximum(row_max, blk_max)
    corr = torch.exp(row_max - new_max)
    p = torch.exp(scores - new_max.unsqueeze(-1))
    acc = acc * corr.unsqueeze(-1) + torch.matmul(p, vj)
    denom = denom * corr + p.sum(dim=-1)
    return acc, new_max, denom


class _RingAttentionFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, group, causal: bool, scale: float):
        world = dist.get_world_size(group) if dist.is_initialized() else 1
        rank = dist.get_rank(group) if dist.is_initialized() else 0
        B, H, Sq, D = q.shape
        qf, kf, vf = q.float(), k.float(), v.float()
        acc = torch.zeros_like(qf)
        row_max = torch.full((B, H, Sq), float("-inf"), device=q.device)
        denom = torch.zeros(B, H, Sq, device=q.device)

        kj, vj = kf, vf
        j = rank
        for step in range(world):
            if (not causal) or j <= rank:
                acc, row_max, denom = _block_update(
                    qf, kj, vj, acc, row_max, denom, scale,
                    local_causal=causal and j == rank)
            if step < world - 1:
                kj, vj = _ring_pass((kj, vj), group)
                j = (j - 1) % world
        out = acc / denom.unsqueeze(-1)
        lse = row_max + denom.log()                 # [B,H,Sq]
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.group = group
        ctx.causal = causal
        ctx.scale = scale
        return out.to(q.dtype)

    @staticmethod
    def backward(ctx, gout):
        q, k, v, out, lse = ctx.saved_tensors
        group, causal, scale = ctx.group, ctx.causal, ctx.scale
        world = dist.get_world_size(group) if dist.is_initialized() else 1
        rank = dist.get_rank(group) if dist.is_initialized() else 0
        qf, gf = q.float(), gout.float()
        Dsum = (gf * out).sum(dim=-1)               # [B,H,Sq]
        dq = torch.zeros_like(qf)

        # (k_j, v_j, dk_j, dv_j) travel the ring together; after `world`
        # hops each block's accumulated gradients return to their owner
        kj, vj = k.float(), v.float()
        dkj = torch.zeros_like(kj)
        dvj = torch.zeros_like(vj)
        j = rank
        for step in range(world):
            if (not causal) or j <= rank:
                scores = torch.matmul(qf, kj.transpose(-1, -2)) * scale
                if causal and j == rank:
                    Sq, Skv = scores.shape[-2], scores.shape[-1]
                    cm = torch.ones(Sq, Skv, dtype=torch.bool,
                                    device=scores.device).tril()
                    scores = scores.masked_fill(~cm, float("-inf"))
                p = torch.exp(scores - lse.unsqueeze(-1))     # true softmax
                dvj = dvj + torch.matmul(p.transpose(-1, -2), gf)
                dp = torch.matmul(gf, vj.transpose(-1, -2))
                ds = p * (dp - Dsum.unsqueeze(-1))
                dq = dq + torch.matmul(ds, kj) * scale
                dkj = dkj + torch.matmul(ds.transpose(-1, -2), qf) * scale
            if step < world - 1:
                kj, vj, dkj, dvj = _ring_pass((kj, vj, dkj, dvj), group)
                j = (j - 1) % world
        if world > 1:
            # one final hop returns each block (and its grads) home
            kj, vj, dkj, dvj = _ring_pass((kj, vj, dkj, dvj), group)
        return (dq.to(q.dtype), dkj.to(k.dtype), dvj.to(v.dtype),
                None, None, None)


def ring_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                   group=None, causal: bool = True,
                   scale: Optional[float] = None) -> torch.Tensor:
    """Exact attention over the ring-sharded global sequence.
    q/k/v: [B, H, S_local, D] (equal head counts — expand GQA KV first)."""
    if scale is None:
        scale = q.shape[-1] ** -0.5
    return _RingAttentionFn.apply(q, k, v, group, causal, scale)
