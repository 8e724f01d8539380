from . import comm
from .zero import ZeroEngine, create_engine

__all__ = ["ZeroEngine", "comm", "create_engine"]
