"""Dedicated training loop for pipeline parallelism (PP and PP x DP).

PP composes differently from the data-parallel Trainer (micro-batches flow
THROUGH ranks instead of being sharded ACROSS them), so `--pp` runs this
compact loop. With world > pp the remaining factor is data parallelism:
ranks lay out as dp_replica x pp_stage (rank = dp_idx * pp + stage), each
pipeline column consumes its own data shard, and stage gradients
all-reduce across the per-stage DP group (flat FlatAdamW buffers -> one
collective per group) before the AdamW step.

The adaptive orchestrator does not attach here (its interventions assume a
whole-model rank); PP x TP/EP mesh composition is a round-2 item
(ROADMAP.md).

Reference capability: vendored ColossalAI pipeline (p2p.py, one_f_one_b.py,
interleaved_pp.py) reached through HybridParallelPlugin (pp_size x dp).
"""

from __future__ import annotations

import os
from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from ..parallel import comm
from .optimizer import FlatAdamW
from .schedulers import WarmupScheduler


def _micro_batches(batch: Dict, n: int) -> List[Dict]:
    """Split one global batch dict into n micro-batch dicts along dim 0."""
    B = batch["input_ids"].shape[0]
    per = max(1, B // n)
    out = []
    for i in range(0, B, per):
        out.append({k: v[i:i + per] for k, v in batch.items()
                    if torch.is_tensor(v)})
    return out


def build_pp_tp_groups(world: int, pp: int, tp: int):
    """Groups for a dp x pp x tp mesh, rank = dp*(pp*tp) + stage*tp + tp_rank
    (tp innermost: a stage's TP shard group is contiguous -> one xGMI hop).
    Returns (pp_group, tp_group, dp_group, dp_idx, stage, tp_rank); every
    rank must call with identical args (dist.new_group is collective).
    Reference capability: ColossalAI HybridParallelPlugin's TPxPPxDP mesh
    (booster/plugin/hybrid_parallel_plugin.py:880)."""
    rank = comm.get_rank()
    assert world % (pp * tp) == 0
    dp = world // (pp * tp)
    pp_groups, tp_groups, dp_groups = {}, {}, {}
    for d in range(dp):
        for t in range(tp):
            pp_groups[(d, t)] = dist.new_group(
                [d * pp * tp + s * tp + t for s in range(pp)])
        for s in range(pp):
            tp_groups[(d, s)] = dist.new_group(
                [d * pp * tp + s * tp + t for t in range(tp)])
    for s in range(pp):
        for t in range(tp):
            dp_groups[(s, t)] = dist.new_group(
                [d * pp * tp + s * tp + t for d in range(dp)])
    dp_idx = rank // (pp * tp)
    stage = (rank // tp) % pp
    tp_rank = rank % tp
    return (pp_groups[(dp_idx, tp_rank)], tp_groups[(dp_idx, stage)],
            dp_groups[(stage, tp_rank)], dp_idx, stage, tp_rank)


class _TPShim:
    """Minimal mesh view for convert_to_tensor_parallel inside a pp x tp
    composition (the global mesh stays unset; groups come from
    build_pp_tp_groups)."""

    def __init__(self, tp_size, tp_rank, tp_group):
        self.tp_size = tp_size
        self.tp_rank = tp_rank
        self.tp_group = tp_group


def build_pp_dp_groups(world: int, pp: int):
    """(pp_group, dp_group, dp_idx, stage) for this rank; every rank must
    call this with identical arguments (dist.new_group is collective)."""
    rank = comm.get_rank()
    dp = world // pp
    pp_groups = [dist.new_group(list(range(d * pp, (d + 1) * pp)))
                 for d in range(dp)]
    dp_groups = [dist.new_group(list(range(s, world, pp)))
                 for s in range(pp)]
    dp_idx, stage = rank // pp, rank % pp
    return pp_groups[dp_idx], dp_groups[stage], dp_idx, stage


def run_pipeline_training(model, cfg, train_ds, logger, pp: int = 0,
                          virtual_stages: int = 1, tp: int = 1,
                          steps: Optional[int] = None) -> Dict:
    """Train `model` over a pp (x tp) (x dp) world. Returns summary stats."""
    from torch.utils.data import DataLoader, Subset

    world = comm.get_world_size()
    rank = comm.get_rank()
    pp = pp or (world // max(tp, 1))
    dp = world // (pp * max(tp, 1))
    pp_group = dp_group = None
    dp_idx, stage = 0, rank
    if tp > 1:
        from ..parallel.tensor_parallel import convert_to_tensor_parallel
        pp_group, tp_group, dp_group, dp_idx, stage, tp_rank = \
            build_pp_tp_groups(world, pp, tp)
        convert_to_tensor_parallel(model, _TPShim(tp, tp_rank, tp_group))
        if dp > 1 and hasattr(train_ds, "__len__"):
            idx = list(range(dp_idx, len(train_ds), dp))
            train_ds = Subset(train_ds, idx)
        if dp == 1:
            dp_group = None
    elif dp > 1:
        pp_group, dp_group, dp_idx, stage = build_pp_dp_groups(world, pp)
        # map-style datasets shard across DP replicas (all stages of one
        # column already share the stream: per-column seed in main.py)
        if hasattr(train_ds, "__len__"):
            idx = list(range(dp_idx, len(train_ds), dp))
            train_ds = Subset(train_ds, idx)

    if virtual_stages > 1:
        from ..parallel.pipeline import InterleavedPipelineEngine
        engine = InterleavedPipelineEngine(model, cfg,
                                           virtual_stages=virtual_stages,
                                           pp_group=pp_group)
        stage_mod = engine.chunks
    else:
        from ..parallel.pipeline import PipelineParallelEngine
        engine = PipelineParallelEngine(model, cfg, pp_group=pp_group)
        stage_mod = engine.stage
    opt = FlatAdamW(stage_mod, lr=cfg.learning_rate,
                    weight_decay=cfg.weight_decay, max_grad_norm=1.0)
    if dp > 1:
        # replicas start identical (same init seed, but broadcast makes it
        # robust for ad-hoc callers)
        src = dist.get_process_group_ranks(dp_group)[0]
        for g in opt.groups:
            dist.broadcast(g.weight_view(), src, group=dp_group)
            if not g._master_is_params:   # refresh the fp32 master copy
                g.master.copy_(g.flat_w[g.shard_lo:g.shard_hi].float())
    accum = max(1, cfg.gradient_accumulation_steps)
    dl = DataLoader(train_ds, batch_size=(cfg.micro_batch_size or 1) * accum,
                    shuffle=False, drop_last=True,
                    num_workers=0)  # every stage consumes the SAME stream
    total = max(1, len(dl)) * max(1, cfg.num_epochs)
    warmup = max(1, int(total * getattr(cfg, "warmup_ratio", 0.1)))
    sched = WarmupScheduler(opt, total_steps=total, warmup_steps=warmup,
                            kind=getattr(cfg, "lr_scheduler", "cosine"))

    global_step = 0
    last_loss = 0.0
    for epoch in range(max(1, cfg.num_epochs)):
        for batch in dl:
            micro = _micro_batches(batch, accum)
            out = engine.train_batch(micro)
            if dp > 1:
                for g in opt.groups:
                    dist.all_reduce(g.flat_g, group=dp_group)
            # mean over micro-batches and DP replicas
            opt.step(grad_scale=1.0 / (len(micro) * dp))
            opt.zero_grad()
            sched.step()
            global_step += 1
            loss = out["loss"]
            if dp > 1:
                loss = loss.clone()
                dist.all_reduce(loss, group=dp_group)
                loss = loss / dp
            last_loss = float(loss)
            if steps is not None and global_step >= steps:
                break
            if global_step % 10 == 0 and stage == pp - 1:
                logger.info(f"pp step {global_step}: loss {last_loss:.4f} "
                            f"lr {opt.groups[0].lr:.2e}")
        if steps is not None and global_step >= steps:
            break

    # per-stage checkpoint: DP replica 0 writes each stage's slice
    exp_dir = os.path.join("experiments", cfg.experiment_name, "checkpoints")
    os.makedirs(exp_dir, exist_ok=True)
    path = os.path.join(exp_dir, f"pp_stage_rank{stage}.pt")
    if dp_idx == 0:
        torch.save({"stage_state_dict": stage_mod.state_dict(),
                    "optimizer_state_dict": opt.state_dict(),
                    "global_step": global_step,
                    "pp_rank": stage, "pp_world": pp, "dp_world": dp,
                    "virtual_stages": virtual_stages}, path)
    if dp > 1:
        dist.barrier()
    return {"global_step": global_step, "loss": last_loss,
            "checkpoint": path, "pp_rank": stage, "dp_idx": dp_idx}
