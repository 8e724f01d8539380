"""Process-group mesh for hybrid DP x EP (x SP) parallelism.

MI355X-native replacement for the reference's ColossalAI ProcessGroupMesh
(/root/reference/Src/Main_Scripts/ColossalAI/colossalai/cluster/
process_group_mesh.py:24) and DeepSpeed expert-parallel group logic
(trainer.py:890-917, backend_deepspeed.py:196). The node is 8 MI355X on
fully-connected xGMI (7 point-to-point links per GPU), so EP all-to-all is
topology-native: every peer pair has a dedicated link.

Layout (world = dp_size x ep_size x tp_size, tp innermost):
    rank = dp_rank * (ep_size * tp_size) + ep_rank * tp_size + tp_rank
- tp_group: contiguous tp ranks; they SHARE a batch shard and hold
  column/row slices of the same weights (parallel/tensor_parallel.py).
- ep_group: ranks of one replica with the same tp_rank (stride tp);
  MoE token all-to-all runs here.
- expert_dp_group: ranks owning the SAME expert (and TP) shard across
  replicas (stride ep*tp); expert grads all-reduce here.
- tp_shard_dp_group: ranks with the same tp_rank across dp x ep; TP
  weight-shard grads of NON-expert params all-reduce here.
- shard_block_group: one replica's full ep x tp block (contiguous);
  sharded-param grad norms sum here (each distinct shard exactly once).
- the full world group is the DP group for replicated params.
TP and EP compose; SP stays exclusive of both.
"""

from __future__ import annotations

from typing import Optional

import torch.distributed as dist

from . import comm

_MESH: Optional["ParallelMesh"] = None


class ParallelMesh:
    def __init__(self, ep_size: int = 1, sp_size: int = 1, tp_size: int = 1,
                 sp_mode: str = "ulysses"):
        world = comm.get_world_size()
        rank = comm.get_rank()
        for name, sz in (("ep", ep_size), ("sp", sp_size), ("tp", tp_size)):
            assert world % max(sz, 1) == 0, \
                f"world {world} not divisible by {name}_size {sz}"
        assert not (sp_size > 1 and (ep_size > 1 or tp_size > 1)), \
            "SP is exclusive of EP/TP in the mesh (TP x EP compose; PPxDP and ZeRO-3+EP compose elsewhere)"
        assert world % (ep_size * tp_size * max(sp_size, 1)) == 0
        self.world = world
        self.rank = rank
        self.ep_size = ep_size
        self.sp_size = sp_size
        # sequence-parallel exchange pattern (reference shardformer
        # SUPPORT_SP_MODE, shard_config.py:13): "ulysses" = head<->sequence
        # all-to-all; "ring" = blockwise ring attention
        # (parallel/context_parallel.py) for sp_size beyond num_kv_heads
        assert sp_mode in ("ulysses", "ring")
        self.sp_mode = sp_mode
        self.tp_size = tp_size
        self.dp_size = world // (ep_size * max(sp_size, 1) * tp_size)
        self.tp_rank = rank % tp_size if tp_size > 1 else 0
        self.ep_rank = (rank // tp_size) % ep_size if ep_size > 1 else 0
        self.dp_rank = rank // (ep_size * tp_size) \
            if (ep_size > 1 or tp_size > 1) else rank
        self.sp_rank = rank % sp_size if sp_size > 1 else 0
        self.ep_group = None
        self.expert_dp_group = None
        self.sp_group = None
        self.tp_group = None
        self.tp_shard_dp_group = None
        self.shard_block_group = None

        def contiguous_groups(size):
            mine = None
            for d in range(world // size):
                ranks = list(range(d * size, (d + 1) * size))
                g = dist.new_group(ranks)
                if rank in ranks:
                    mine = g
            return mine

        def strided_groups(size):
            mine = None
            for e in range(size):
                ranks = list(range(e, world, size))
                g = dist.new_group(ranks)
                if rank in ranks:
                    mine = g
            return mine

        def grid_groups(stride, size):
            """groups {base + i*stride : i < size} tiled over the world."""
            mine = None
            block = stride * size
            for b in range(0, world, block):
                for off in range(stride):
                    ranks = [b + off + i * stride for i in range(size)]
                    g = dist.new_group(ranks)
                    if rank in ranks:
                        mine = g
            return mine

        if world > 1 and ep_size > 1:
            # stride tp within each dp block (== contiguous when tp == 1)
            self.ep_group = grid_groups(tp_size, ep_size)
            self.expert_dp_group = strided_groups(ep_size * tp_size)
        if world > 1 and sp_size > 1:
            self.sp_group = contiguous_groups(sp_size)
        if world > 1 and tp_size > 1:
            self.tp_group = contiguous_groups(tp_size)
            self.tp_shard_dp_group = strided_groups(tp_size)
        if world > 1 and ep_size > 1 and tp_size > 1:
            self.shard_block_group = contiguous_groups(ep_size * tp_size)

    @property
    def expert_dp_size(self) -> int:
        return self.world // self.ep_size if self.ep_size > 1 else self.world

    # ---- generic "sharded param" accessors (EP expert shards and TP weight
    # shards use the same engine machinery: reduce grads across the replica
    # group; sum norms across one exchange group).  Two classes:
    #   "expert": EP-sharded expert params (under TP x EP also TP-sharded);
    #             replicas live across dp only.
    #   "tp":     TP-sharded non-expert params; replicas across dp x ep.
    def replica_group_for(self, comm: str):
        if comm == "expert" and self.ep_size > 1:
            return self.expert_dp_group
        return self.tp_shard_dp_group if self.tp_size > 1 \
            else self.expert_dp_group

    def replica_size_for(self, comm: str) -> int:
        if comm == "expert" and self.ep_size > 1:
            return self.dp_size
        if self.tp_size > 1:
            return self.world // self.tp_size
        return self.dp_size if self.ep_size > 1 else 1

    def exchange_group_for(self, comm: str):
        if comm == "expert" and self.ep_size > 1:
            return self.shard_block_group if self.tp_size > 1 \
                else self.ep_group
        return self.tp_group if self.tp_size > 1 else self.ep_group

    def exchange_size_for(self, comm: str) -> int:
        if comm == "expert" and self.ep_size > 1:
            return self.ep_size * self.tp_size
        return self.tp_size if self.tp_size > 1 else self.ep_size

    # legacy single-class views (exclusive meshes; zero3 uses these)
    @property
    def shard_replica_group(self):
        if self.tp_size > 1:
            return self.tp_shard_dp_group
        return self.expert_dp_group

    @property
    def shard_replica_size(self) -> int:
        if self.tp_size > 1:
            return self.world // self.tp_size
        if self.ep_size > 1:
            return self.world // self.ep_size
        return 1

    @property
    def shard_exchange_group(self):
        if self.tp_size > 1:
            return self.tp_group
        return self.ep_group

    @property
    def shard_exchange_size(self) -> int:
        return self.tp_size if self.tp_size > 1 else self.ep_size


def init_mesh(ep_size: int = 1, sp_size: int = 1,
              tp_size: int = 1, sp_mode: str = "ulysses") -> ParallelMesh:
    global _MESH
    _MESH = ParallelMesh(ep_size, sp_size, tp_size, sp_mode=sp_mode)
    return _MESH


def get_mesh() -> Optional[ParallelMesh]:
    return _MESH


def reset_mesh():
    global _MESH
    _MESH = None
