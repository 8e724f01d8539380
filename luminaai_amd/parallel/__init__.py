from . import comm
from .expert_parallel import all_to_all, is_expert_param, sync_expert_grads
from .mesh import ParallelMesh, get_mesh, init_mesh, reset_mesh
from .zero import ZeroEngine, create_engine

__all__ = ["ParallelMesh", "ZeroEngine", "all_to_all", "comm",
           "create_engine", "get_mesh", "init_mesh", "is_expert_param",
           "reset_mesh", "sync_expert_grads"]
