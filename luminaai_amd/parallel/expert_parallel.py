"""Expert-parallel all-to-all primitives.

MI355X-native equivalent of the reference's EP machinery (DeepSpeed MoE
all-to-all dispatch, trainer.py:842-843; ColossalAI moe/_operation.py:105
AllToAll). On an 8-GPU xGMI node all-to-all is the topology-native
collective: each of the 7 peers is reached over its own ~153 GB/s link, so
the token exchange runs at near-full bisection — unlike ring all-reduce,
which is single-link bound.

The exchanged buffers are capacity-bucketed and therefore STATIC-SHAPED
([ep, E_local, C, h]) — no length metadata exchange, no host syncs, and the
pattern is hipGraph-capturable.
"""

from __future__ import annotations

import torch
import torch.distributed as dist


class _AllToAllFn(torch.autograd.Function):
    """Differentiable all_to_all_single on dim 0; backward is the inverse
    exchange (all-to-all is self-adjoint under transposition of peers)."""

    @staticmethod
    def forward(ctx, x: torch.Tensor, group):
        ctx.group = group
        x = x.contiguous()
        # NOTE empty_like would PRESERVE non-contiguous strides and the
        # collective writes assuming dense layout — always allocate dense.
        out = torch.empty(x.shape, dtype=x.dtype, device=x.device)
        dist.all_to_all_single(out, x, group=group)
        return out

    @staticmethod
    def backward(ctx, gy: torch.Tensor):
        gy = gy.contiguous()
        gx = torch.empty(gy.shape, dtype=gy.dtype, device=gy.device)
        dist.all_to_all_single(gx, gy, group=ctx.group)
        return gx, None


def all_to_all(x: torch.Tensor, group=None) -> torch.Tensor:
    """Exchange equal chunks of dim 0 across the group (autograd-aware)."""
    if group is None and not dist.is_initialized():
        return x
    world = dist.get_world_size(group)
    if world == 1:
        return x
    assert x.shape[0] % world == 0, \
        f"dim0 {x.shape[0]} not divisible by group size {world}"
    return _AllToAllFn.apply(x, group)


def is_expert_param(name: str) -> bool:
    """Parameters sharded along EP (not DP-replicated)."""
    return ".w_gate_up" in name or ".w_down" in name


def sync_expert_grads(model, expert_dp_group):
    """All-reduce expert-weight grads across replicas owning the same shard
    (no-op when expert_dp_size == 1). Standalone utility for custom
    training loops; the ZeRO engines do this on their flat buffers
    (zero.py reduce_gradients / zero3.py expert segments) instead."""
    if expert_dp_group is None:
        return
    for name, p in model.named_parameters():
        if is_expert_param(name) and p.grad is not None:
            dist.all_reduce(p.grad, group=expert_dp_group)
