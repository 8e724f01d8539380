from . import comm
from .expert_parallel import all_to_all, is_expert_param, sync_expert_grads
from .mesh import ParallelMesh, get_mesh, init_mesh, reset_mesh
from .sequence_parallel import (scatter_heads_gather_seq,
                                scatter_seq_gather_heads, shard_sequence)
from .zero import ZeroEngine, create_engine

__all__ = ["ParallelMesh", "ZeroEngine", "all_to_all", "comm",
           "create_engine", "get_mesh", "init_mesh", "is_expert_param",
           "reset_mesh", "scatter_heads_gather_seq",
           "scatter_seq_gather_heads", "shard_sequence", "sync_expert_grads"]
