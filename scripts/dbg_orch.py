import json, os, sys, time
import torch
sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
from luminaai_amd.config import ConfigPresets
from luminaai_amd.data.dataset import SyntheticDataset
from luminaai_amd.data.tokenizer import ConversationTokenizer
from luminaai_amd.models import DeepSeekTransformer, config_to_deepseek_config
from luminaai_amd.training import AdaptiveTrainingOrchestrator, Trainer

def run(mode, steps=30):
    cfg = ConfigPresets.debug(); cfg.num_workers=0; cfg.micro_batch_size=4
    cfg.gradient_accumulation_steps=1; cfg.eval_every_n_batches=0; cfg.save_every_n_batches=0
    cfg.enable_adaptive_lr = True
    torch.manual_seed(0)
    model = DeepSeekTransformer(config_to_deepseek_config(cfg))
    tok = ConversationTokenizer(max_length=cfg.seq_length)
    t = Trainer(model, tok, cfg); t._setup_scheduler(steps+10)
    orch=None
    if mode == "noop_hook":
        t.set_metrics_hook(lambda m: None)
    if mode == "noop_nostage":
        t.set_metrics_hook(lambda m: None)
        t.get_current_metrics = lambda *a, **k: None   # isolate staging
    if mode in ("hook_only", "thread_only", "full"):
        orch = AdaptiveTrainingOrchestrator(cfg, trainer=t)
        orch.initialize_training()
        if mode in ("thread_only", "full"):
            orch.start_real_time_monitoring()
        if mode == "thread_only":
            t.set_metrics_hook(None)     # thread runs, nothing enqueued
    ds = SyntheticDataset(cfg.vocab_size, cfg.seq_length, 4*steps, seed=1)
    for i in range(3):
        ids = torch.stack([ds[j]["input_ids"] for j in range(4)]); lab = torch.stack([ds[j]["labels"] for j in range(4)])
        t.engine.set_sync(True); t.train_step({"input_ids": ids, "labels": lab}); t.optimizer_step()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for i in range(steps):
        ids = torch.stack([ds[4*i+j]["input_ids"] for j in range(4)]); lab = torch.stack([ds[4*i+j]["labels"] for j in range(4)])
        t.engine.set_sync(True); out=t.train_step({"input_ids": ids, "labels": lab}); t.optimizer_step(); t._emit_metrics(out)
    torch.cuda.synchronize(); dt=time.perf_counter()-t0
    if orch: orch.cleanup()
    return steps/dt

for mode in ("plain", "noop_hook", "noop_nostage", "hook_only", "noop_hook",
             "plain", "hook_only"):
    print(mode, round(run(mode), 2))
