"""Process-group mesh for hybrid DP x EP (x SP) parallelism.

MI355X-native replacement for the reference's ColossalAI ProcessGroupMesh
(/root/reference/Src/Main_Scripts/ColossalAI/colossalai/cluster/
process_group_mesh.py:24) and DeepSpeed expert-parallel group logic
(trainer.py:890-917, backend_deepspeed.py:196). The node is 8 MI355X on
fully-connected xGMI (7 point-to-point links per GPU), so EP all-to-all is
topology-native: every peer pair has a dedicated link.

Layout (world = dp_size x ep_size, ep contiguous):
    rank = dp_rank * ep_size + ep_rank
- ep_group: ranks sharing a data-parallel replica, experts sharded over them;
  MoE token all-to-all runs here.
- expert_dp_group: ranks owning the SAME expert shard across replicas
  (stride ep_size); expert grads all-reduce here.
- the full world group is the DP group for non-expert (replicated) params.
"""

from __future__ import annotations

from typing import Optional

import torch.distributed as dist

from . import comm

_MESH: Optional["ParallelMesh"] = None


class ParallelMesh:
    def __init__(self, ep_size: int = 1, sp_size: int = 1):
        world = comm.get_world_size()
        rank = comm.get_rank()
        assert world % max(ep_size, 1) == 0, \
            f"world {world} not divisible by ep_size {ep_size}"
        assert world % max(sp_size, 1) == 0, \
            f"world {world} not divisible by sp_size {sp_size}"
        assert ep_size <= 1 or sp_size <= 1, \
            "EP and Ulysses-SP composition is not supported yet"
        self.world = world
        self.rank = rank
        self.ep_size = ep_size
        self.sp_size = sp_size
        self.dp_size = world // (ep_size * max(sp_size, 1))
        self.ep_rank = rank % ep_size if ep_size > 1 else 0
        self.dp_rank = rank // ep_size
        self.sp_rank = rank % sp_size if sp_size > 1 else 0
        self.ep_group = None
        self.expert_dp_group = None
        self.sp_group = None
        if world > 1 and ep_size > 1:
            # build ALL groups on every rank (dist.new_group is collective)
            for d in range(world // ep_size):
                ranks = list(range(d * ep_size, (d + 1) * ep_size))
                g = dist.new_group(ranks)
                if rank in ranks:
                    self.ep_group = g
            for e in range(ep_size):
                ranks = list(range(e, world, ep_size))
                g = dist.new_group(ranks)
                if rank in ranks:
                    self.expert_dp_group = g
        if world > 1 and sp_size > 1:
            for d in range(world // sp_size):
                ranks = list(range(d * sp_size, (d + 1) * sp_size))
                g = dist.new_group(ranks)
                if rank in ranks:
                    self.sp_group = g

    @property
    def expert_dp_size(self) -> int:
        return self.world // self.ep_size if self.ep_size > 1 else self.world


def init_mesh(ep_size: int = 1, sp_size: int = 1) -> ParallelMesh:
    global _MESH
    _MESH = ParallelMesh(ep_size, sp_size)
    return _MESH


def get_mesh() -> Optional[ParallelMesh]:
    return _MESH


def reset_mesh():
    global _MESH
    _MESH = None
