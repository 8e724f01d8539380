#!/usr/bin/env python3
"""Orchestrator overhead: steps/s with vs without the adaptive control plane
(reference Preformance_Overhead.md claimed 3-8% at 1-2 GPUs)."""

from __future__ import annotations

import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from luminaai_amd.config import ConfigPresets  # noqa: E402
from luminaai_amd.data.dataset import SyntheticDataset  # noqa: E402
from luminaai_amd.data.tokenizer import ConversationTokenizer  # noqa: E402
from luminaai_amd.models import DeepSeekTransformer, config_to_deepseek_config  # noqa: E402
from luminaai_amd.training import AdaptiveTrainingOrchestrator, Trainer  # noqa: E402


def run(with_orchestrator: bool, steps: int = 30) -> float:
    cfg = ConfigPresets.debug()
    cfg.num_workers = 0
    cfg.micro_batch_size = 4
    cfg.gradient_accumulation_steps = 1
    cfg.eval_every_n_batches = 0
    cfg.save_every_n_batches = 0
    cfg.enable_adaptive_lr = True
    torch.manual_seed(0)
    model = DeepSeekTransformer(config_to_deepseek_config(cfg))
    tok = ConversationTokenizer(max_length=cfg.seq_length)
    trainer = Trainer(model, tok, cfg)
    trainer._setup_scheduler(steps + 10)
    orch = None
    if with_orchestrator:
        orch = AdaptiveTrainingOrchestrator(cfg, trainer=trainer)
        orch.initialize_training()
        orch.start_real_time_monitoring()
    ds = SyntheticDataset(cfg.vocab_size, cfg.seq_length, 4 * steps, seed=1)
    # warmup
    for i in range(3):
        ids = torch.stack([ds[j]["input_ids"] for j in range(4)])
        lab = torch.stack([ds[j]["labels"] for j in range(4)])
        trainer.engine.set_sync(True)
        trainer.train_step({"input_ids": ids, "labels": lab})
        trainer.optimizer_step()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(steps):
        base = 4 * i
        ids = torch.stack([ds[base + j]["input_ids"] for j in range(4)])
        lab = torch.stack([ds[base + j]["labels"] for j in range(4)])
        trainer.engine.set_sync(True)
        out = trainer.train_step({"input_ids": ids, "labels": lab})
        trainer.optimizer_step()
        trainer._emit_metrics(out)
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    if orch is not None:
        orch.cleanup()
    return steps / dt


def main():
    # In-process run order matters at 4 ms debug-scale steps: the SECOND
    # model build in a process measures ~35% slower regardless of mode
    # (allocator growth; verified with no-op hooks in scripts/dbg_orch.py).
    # Discard a warmup run and interleave the arms, reporting medians.
    run(False, steps=10)                      # discarded allocator warmup
    plains, adaptives = [], []
    for _ in range(3):
        plains.append(run(False))
        adaptives.append(run(True))
    plains.sort()
    adaptives.sort()
    plain, adaptive = plains[1], adaptives[1]
    res = {
        "steps_per_sec_plain": round(plain, 3),
        "steps_per_sec_adaptive": round(adaptive, 3),
        "orchestrator_overhead_pct": round((plain - adaptive) / plain * 100, 2),
        "runs_plain": [round(x, 1) for x in plains],
        "runs_adaptive": [round(x, 1) for x in adaptives],
    }
    print(json.dumps(res))


if __name__ == "__main__":
    main()
