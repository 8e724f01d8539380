"""Tensor parallelism: Megatron-style column/row-parallel dense blocks.

MI355X-native rebuild of the capability the reference reached through
vendored ColossalAI shardformer (layer/linear.py,
qkv_fused_linear.py, _operation.py column/row parallel matmul). Scope:

- attention: head-aligned shard of the fused qkv projection (column
  parallel) + o_proj row parallel with output all-reduce;
- dense SwiGLU: gate/up halves sharded along the intermediate dim (column)
  + down projection row parallel with output all-reduce;
- embeddings / norms / lm_head / MoE layers stay replicated (MoE layers
  behave as plain DP inside the TP group — EP is the memory-scaling story
  for experts; composition is round-2);
- f/g conjugate functions: `tp_copy` (identity fwd, grad all-reduce bwd) at
  block inputs, `tp_reduce` (all-reduce fwd, identity bwd) at block outputs.

On xGMI the per-layer activation all-reduce makes TP the least-preferred
strategy (PARITY.md) — this exists for capability parity and for layers
whose shards exceed one GPU even under ZeRO-3+EP.

Sharded weights are tagged `_shard_parallel`; the optimizer places them in
the sharded comm group (grads all-reduced over mesh.shard_replica_group,
never the global DP group) — same machinery as EP expert shards.
"""

from __future__ import annotations

import torch
import torch.distributed as dist
import torch.nn as nn


class _TPCopyFn(torch.autograd.Function):
    """f: identity forward; backward all-reduces the input grad across TP
    (each TP rank back-propagates a partial through its weight shard)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return x

    @staticmethod
    def backward(ctx, gy):
        gy = gy.contiguous()
        dist.all_reduce(gy, group=ctx.group)
        return gy, None


class _TPReduceFn(torch.autograd.Function):
    """g: all-reduce forward (sum of row-parallel partials); identity bwd."""

    @staticmethod
    def forward(ctx, x, group):
        x = x.contiguous()
        dist.all_reduce(x, group=group)
        return x

    @staticmethod
    def backward(ctx, gy):
        return gy, None


def tp_copy(x, group):
    return _TPCopyFn.apply(x, group) if group is not None else x


def tp_reduce(x, group):
    return _TPReduceFn.apply(x, group) if group is not None else x


def _replace_linear(module: nn.Module, name: str, weight: torch.Tensor):
    old = getattr(module, name)
    new = nn.Linear(weight.shape[1], weight.shape[0], bias=False,
                    device=old.weight.device, dtype=old.weight.dtype)
    with torch.no_grad():
        new.weight.copy_(weight)
    new.weight._shard_parallel = True
    setattr(module, name, new)


@torch.no_grad()
def convert_to_tensor_parallel(model, mesh) -> int:
    """Shard the dense blocks of a DeepSeekTransformer across mesh.tp_size
    ranks IN PLACE (each rank keeps its slice of the full weights the model
    was constructed with — construct with identical seeds on all ranks).
    Returns the number of converted layers."""
    tp = mesh.tp_size
    if tp <= 1:
        return 0
    r = mesh.tp_rank
    group = mesh.tp_group
    n = 0
    for layer in model.layers:
        attn = layer.attention
        assert attn.num_heads % tp == 0 and attn.num_kv_heads % tp == 0, \
            "num_heads and num_kv_heads must divide tp_size"
        hd = attn.head_dim
        qs, kvs = attn.q_size, attn.kv_size
        W = attn.qkv_proj.weight.data
        h_in = W.shape[1]
        lh = attn.num_heads // tp
        lkv = attn.num_kv_heads // tp
        qw = W[:qs].view(attn.num_heads, hd, h_in)[r * lh:(r + 1) * lh]
        kw = W[qs:qs + kvs].view(attn.num_kv_heads, hd, h_in)[
            r * lkv:(r + 1) * lkv]
        vw = W[qs + kvs:].view(attn.num_kv_heads, hd, h_in)[
            r * lkv:(r + 1) * lkv]
        _replace_linear(attn, "qkv_proj",
                        torch.cat([qw.reshape(-1, h_in), kw.reshape(-1, h_in),
                                   vw.reshape(-1, h_in)]))
        ow = attn.o_proj.weight.data           # [h, q_size]
        ow_shard = ow.view(h_in, attn.num_heads, hd)[:, r * lh:(r + 1) * lh] \
            .reshape(h_in, lh * hd)
        _replace_linear(attn, "o_proj", ow_shard)
        attn.num_heads = lh
        attn.num_kv_heads = lkv
        attn.q_size = lh * hd
        attn.kv_size = lkv * hd
        attn.tp_group = group

        ffn = layer.ffn
        if hasattr(ffn, "gate_up_proj"):       # dense SwiGLU
            I = ffn.intermediate_size
            assert I % tp == 0
            li = I // tp
            gu = ffn.gate_up_proj.weight.data  # [2I, h]
            gshard = gu[:I][r * li:(r + 1) * li]
            ushard = gu[I:][r * li:(r + 1) * li]
            _replace_linear(ffn, "gate_up_proj", torch.cat([gshard, ushard]))
            dw = ffn.down_proj.weight.data     # [h, I]
            _replace_linear(ffn, "down_proj", dw[:, r * li:(r + 1) * li])
            ffn.intermediate_size = li
            ffn.tp_group = group
        elif hasattr(ffn, "w_gate_up"):        # MoE expert stack
            # Megatron-style TP over the batched expert weights (reference
            # capability: ColossalAI hybrid_parallel_plugin.py:880 /
            # moe_hybrid_parallel_plugin.py:92): gate/up column-parallel on
            # the intermediate dim, down row-parallel; the partial expert
            # outputs all-reduce across the TP group inside the layer
            # forward (transformer.py MoEFFNLayer._mlp).
            I = ffn.intermediate_size
            assert I % tp == 0
            li = I // tp
            gu = ffn.w_gate_up.data            # [EL, h, 2I]
            gslice = gu[:, :, r * li:(r + 1) * li]
            uslice = gu[:, :, I + r * li:I + (r + 1) * li]
            wg = torch.cat([gslice, uslice], dim=2).contiguous()
            dn = ffn.w_down.data[:, r * li:(r + 1) * li, :].contiguous()
            ffn.w_gate_up = nn.Parameter(wg)
            ffn.w_down = nn.Parameter(dn)
            ffn.w_gate_up._shard_parallel = True
            ffn.w_down._shard_parallel = True
            ffn.intermediate_size = li
            ffn.tp_group = group
            # routing must be IDENTICAL on every TP rank: give the layer a
            # shared-seed generator for its gating noise
            seed = torch.tensor([torch.initial_seed() % (2 ** 31)],
                                dtype=torch.int64)
            if dist.is_initialized():
                seed = seed.to("cuda" if dist.get_backend(group) == "nccl"
                               else "cpu")
                dist.broadcast(seed, src=dist.get_process_group_ranks(group)[0],
                               group=group)
            g = torch.Generator(device=ffn.w_gate_up.device)
            g.manual_seed(int(seed.item()))
            ffn._routing_gen = g
        n += 1
    return n
