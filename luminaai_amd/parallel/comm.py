"""Distributed communication setup: RCCL over xGMI.

Replaces the reference's three wrapper stacks (FSDP/DeepSpeed/ColossalAI NCCL
init) with one torch.distributed process-group layer.
On ROCm the "nccl" backend IS RCCL; intra-node transport is xGMI
(7 point-to-point links x ~153 GB/s per GPU, fully connected 8-GPU node).
Consequences encoded here and in zero.py/ep.py:
- prefer several in-flight medium buckets over one giant bucket (ring
  collectives are per-link bound);
- EP token all-to-all and Ulysses SP are topology-native (7 peers <-> 7 links).
"""

from __future__ import annotations

import datetime
import os
from typing import Optional

import torch
import torch.distributed as dist


def env_rank() -> int:
    return int(os.environ.get("RANK", 0))


def env_local_rank() -> int:
    return int(os.environ.get("LOCAL_RANK", 0))


def env_world_size() -> int:
    return int(os.environ.get("WORLD_SIZE", 1))


def init_distributed(backend: Optional[str] = None,
                     timeout_s: int = 600) -> bool:
    """Initialise torch.distributed from torchrun env vars. Returns True if a
    process group (world > 1 or explicit env) is active."""
    if dist.is_initialized():
        return True
    if env_world_size() <= 1 and "MASTER_ADDR" not in os.environ:
        return False
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    if torch.cuda.is_available():
        torch.cuda.set_device(env_local_rank())
    dist.init_process_group(backend=backend, init_method="env://",
                            timeout=datetime.timedelta(seconds=timeout_s))
    return True


def is_distributed() -> bool:
    return dist.is_initialized() and dist.get_world_size() > 1


def get_rank() -> int:
    return dist.get_rank() if dist.is_initialized() else 0


def get_world_size() -> int:
    return dist.get_world_size() if dist.is_initialized() else 1


def barrier():
    if dist.is_initialized():
        dist.barrier()


def cleanup():
    if dist.is_initialized():
        dist.destroy_process_group()


def broadcast_object(obj, src: int = 0):
    if not is_distributed():
        return obj
    holder = [obj]
    dist.broadcast_object_list(holder, src=src)
    return holder[0]


def all_reduce_scalar(value: float, op: str = "sum") -> float:
    if not is_distributed():
        return value
    t = torch.tensor([value], dtype=torch.float64)
    if torch.cuda.is_available():
        t = t.cuda()
    dist.all_reduce(t, op=dist.ReduceOp.MAX if op == "max" else dist.ReduceOp.SUM)
    return float(t.item())
