from . import comm
from .context_parallel import ring_attention
from .expert_parallel import all_to_all, is_expert_param, sync_expert_grads
from .mesh import ParallelMesh, get_mesh, init_mesh, reset_mesh
from .pipeline import (InterleavedPipelineEngine, PipelineParallelEngine,
                       PipelineStage, partition_layers)
from .sequence_parallel import (scatter_heads_gather_seq,
                                scatter_seq_gather_heads, shard_sequence)
from .tensor_parallel import convert_to_tensor_parallel, tp_copy, tp_reduce
from .zero import ZeroEngine, create_engine
from .zero3 import Zero3Engine

__all__ = ["InterleavedPipelineEngine", "ParallelMesh",
           "PipelineParallelEngine", "PipelineStage",
           "Zero3Engine", "ZeroEngine", "all_to_all", "comm",
           "convert_to_tensor_parallel", "create_engine", "get_mesh",
           "init_mesh", "is_expert_param", "partition_layers", "reset_mesh", "ring_attention",
           "scatter_heads_gather_seq", "scatter_seq_gather_heads",
           "shard_sequence", "sync_expert_grads", "tp_copy", "tp_reduce"]
