"""Ulysses-style sequence parallelism: head<->sequence all-to-all.

MI355X-native rebuild of the capability the reference reached through
vendored ColossalAI's "all_to_all" SP mode (shardformer/layer/_operation.py:
784-930 `_all_to_all`): each rank holds a 1/sp slice of the sequence; before
attention the heads are scattered and the sequence gathered (every head sees
the full context), and the inverse exchange runs after attention. On the
8-GPU xGMI node this is two all-to-alls per layer over the 7 fully-connected
links — the topology-native way to scale context length (SURVEY.md §5
long-context gap).

Token-wise ops (RMSNorm, FFN/MoE, losses) run on the local shard untouched;
RoPE is applied BEFORE the exchange using each shard's global positions.
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from .expert_parallel import _AllToAllFn


def _a2a(x: torch.Tensor, group) -> torch.Tensor:
    return _AllToAllFn.apply(x, group)


def scatter_heads_gather_seq(x: torch.Tensor, sp: int, group) -> torch.Tensor:
    """[B, S_loc, H, D] -> [B, S_loc*sp, H/sp, D] (full sequence, head shard).

    Differentiable; the backward is the inverse exchange.
    """
    B, S, H, D = x.shape
    assert H % sp == 0, f"heads {H} not divisible by sp {sp}"
    Hs = H // sp
    # [B,S,H,D] -> [sp, B, S, Hs, D]: chunk p carries heads for rank p
    xs = x.view(B, S, sp, Hs, D).permute(2, 0, 1, 3, 4).contiguous()
    out = _a2a(xs, group)               # chunk p now: rank p's seq, my heads
    # [sp, B, S, Hs, D] -> [B, sp*S, Hs, D]
    return out.permute(1, 0, 2, 3, 4).reshape(B, sp * S, Hs, D)


def scatter_seq_gather_heads(x: torch.Tensor, sp: int, group) -> torch.Tensor:
    """[B, S_full, H/sp, D] -> [B, S_full/sp, H, D] (inverse exchange)."""
    B, Sf, Hs, D = x.shape
    assert Sf % sp == 0
    S = Sf // sp
    # [B, sp, S, Hs, D] -> [sp, B, S, Hs, D]: chunk p = rank p's seq slice
    xs = x.view(B, sp, S, Hs, D).permute(1, 0, 2, 3, 4).contiguous()
    out = _a2a(xs, group)               # chunk p = my seq slice, rank p heads
    # [sp, B, S, Hs, D] -> [B, S, sp*Hs, D]
    return out.permute(1, 2, 0, 3, 4).reshape(B, S, sp * Hs, D)


def shard_sequence(t: torch.Tensor, sp_rank: int, sp: int,
                   dim: int = 1) -> torch.Tensor:
    """Slice a full-sequence tensor into this rank's contiguous shard."""
    S = t.shape[dim]
    assert S % sp == 0, f"seq {S} not divisible by sp {sp}"
    loc = S // sp
    return t.narrow(dim, sp_rank * loc, loc)
