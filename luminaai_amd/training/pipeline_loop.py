"""Dedicated training loop for pipeline parallelism.

PP composes differently from the data-parallel Trainer (micro-batches flow
THROUGH ranks instead of being sharded ACROSS them), so `--pp` runs this
compact loop: partition the model into stages (1F1B) or pp x v chunks
(interleaved), step a flat fused AdamW over the local stage's parameters,
and checkpoint per stage. The adaptive orchestrator does not attach here
(its interventions assume a whole-model rank); full PP x DP mesh
composition is a round-2 item (ROADMAP.md).

Reference capability: vendored ColossalAI pipeline (p2p.py, one_f_one_b.py,
interleaved_pp.py) reached through HybridParallelPlugin.
"""

from __future__ import annotations

import os
from typing import Dict, List, Optional

import torch

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


def run_pipeline_training(model, cfg, train_ds, logger,
                          virtual_stages: int = 1,
                          steps: Optional[int] = None) -> Dict:
    """Train `model` over the whole PP world. Returns summary stats."""
    from torch.utils.data import DataLoader

    if virtual_stages > 1:
        from ..parallel.pipeline import InterleavedPipelineEngine
        engine = InterleavedPipelineEngine(model, cfg,
                                           virtual_stages=virtual_stages)
        stage_mod = engine.chunks
    else:
        from ..parallel.pipeline import PipelineParallelEngine
        engine = PipelineParallelEngine(model, cfg)
        stage_mod = engine.stage
    rank = comm.get_rank()
    opt = FlatAdamW(stage_mod, lr=cfg.learning_rate,
                    weight_decay=cfg.weight_decay, max_grad_norm=1.0)
    accum = max(1, cfg.gradient_accumulation_steps)
    dl = DataLoader(train_ds, batch_size=(cfg.micro_batch_size or 1) * accum,
                    shuffle=False, drop_last=True,
                    num_workers=0)  # every stage consumes the SAME stream
    total = len(dl) * max(1, cfg.num_epochs)
    warmup = max(1, int(total * getattr(cfg, "warmup_ratio", 0.1)))
    sched = WarmupScheduler(opt, total_steps=total, warmup_steps=warmup,
                            kind=getattr(cfg, "lr_scheduler", "cosine"))

    global_step = 0
    last_loss = 0.0
    for epoch in range(max(1, cfg.num_epochs)):
        for batch in dl:
            micro = _micro_batches(batch, accum)
            out = engine.train_batch(micro)
            # mean over micro-batches; scale grads accordingly
            opt.step(grad_scale=1.0 / len(micro))
            opt.zero_grad()
            sched.step()
            global_step += 1
            last_loss = float(out["loss"])
            if steps is not None and global_step >= steps:
                break
            if global_step % 10 == 0 and engine.pp - 1 == rank:
                logger.info(f"pp step {global_step}: loss {last_loss:.4f} "
                            f"lr {opt.groups[0].lr:.2e}")
        if steps is not None and global_step >= steps:
            break

    # per-stage checkpoint: stage_rank{r}.pt holds this rank's slice
    exp_dir = os.path.join("experiments", cfg.experiment_name, "checkpoints")
    os.makedirs(exp_dir, exist_ok=True)
    path = os.path.join(exp_dir, f"pp_stage_rank{rank}.pt")
    torch.save({"stage_state_dict": stage_mod.state_dict(),
                "optimizer_state_dict": opt.state_dict(),
                "global_step": global_step,
                "pp_rank": rank, "pp_world": engine.pp,
                "virtual_stages": virtual_stages}, path)
    return {"global_step": global_step, "loss": last_loss,
            "checkpoint": path, "pp_rank": rank}
