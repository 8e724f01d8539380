"""hybrid_70b memory-plan validation in reduced-layer form.

Builds the hybrid_70b preset (64-expert top-2 MoE + MoD, h=2048) with a
reduced layer count on ONE GPU, runs two optimizer steps, and reports the
measured memory so the full-model ZeRO-3 x 8-GPU plan in config.py's
estimator can be checked against reality.

Usage: python scripts/hybrid70b_probe.py [--layers 4]
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from luminaai_amd.config import ConfigPresets
from luminaai_amd.data.tokenizer import ConversationTokenizer


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--layers", type=int, default=4)
    args = ap.parse_args()

    cfg = ConfigPresets.hybrid_70b()
    full_layers = cfg.num_layers
    cfg.num_layers = args.layers
    cfg.micro_batch_size = 2
    cfg.gradient_accumulation_steps = 1
    cfg.zero_stage = 0          # single-GPU probe: flat optimizer
    cfg.num_workers = 0
    cfg.eval_every_n_batches = 0
    cfg.save_every_n_batches = 0
    est = cfg.estimate_memory_gb() if hasattr(cfg, "estimate_memory_gb") else None

    from luminaai_amd.models import DeepSeekTransformer, config_to_deepseek_config
    from luminaai_amd.training import Trainer
    torch.manual_seed(0)
    dev = torch.device("cuda")
    with torch.device(dev):
        model = DeepSeekTransformer(config_to_deepseek_config(cfg))
    n_params = sum(p.numel() for p in model.parameters())
    tok = ConversationTokenizer(max_length=cfg.seq_length)
    trainer = Trainer(model, tok, cfg)
    trainer._setup_scheduler(10)

    torch.cuda.reset_peak_memory_stats()
    for i in range(2):
        ids = torch.randint(1, cfg.vocab_size,
                            (cfg.micro_batch_size, cfg.seq_length + 1),
                            device=dev)
        trainer.engine.set_sync(True)
        t0 = time.time()
        trainer.train_step({"input_ids": ids[:, :-1], "labels": ids[:, 1:]})
        trainer.optimizer_step()
        torch.cuda.synchronize()
        print(f"step {i}: {time.time()-t0:.2f}s", flush=True)
    peak = torch.cuda.max_memory_allocated() / 2**30
    per_layer = n_params / args.layers
    print(f"hybrid_70b probe: layers={args.layers}/{full_layers} "
          f"params={n_params/1e9:.2f}B (~{per_layer*full_layers/1e9:.0f}B full) "
          f"peak_mem={peak:.1f} GiB "
          f"(~{peak/args.layers:.2f} GiB/layer at mb2 ckpt={cfg.gradient_checkpointing})",
          flush=True)
    if est is not None:
        print(f"config estimator said: {est}")


if __name__ == "__main__":
    main()
