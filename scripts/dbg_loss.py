import os, sys, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from luminaai_amd.config import ConfigPresets
from luminaai_amd.data.dataset import SyntheticDataset
from luminaai_amd.data.tokenizer import ConversationTokenizer
from luminaai_amd.models import DeepSeekTransformer, config_to_deepseek_config
from luminaai_amd.training import Trainer

def run(total_sched, steps=8):
    cfg = ConfigPresets.b1_moe()
    cfg.num_workers = 0; cfg.micro_batch_size = 8
    cfg.gradient_accumulation_steps = 1
    cfg.gradient_checkpointing = False
    cfg.eval_every_n_batches = 0; cfg.save_every_n_batches = 0
    torch.manual_seed(cfg.seed)
    dev = torch.device("cuda")
    with torch.device(dev):
        model = DeepSeekTransformer(config_to_deepseek_config(cfg))
    t = Trainer(model, ConversationTokenizer(max_length=cfg.seq_length), cfg)
    t._setup_scheduler(total_sched)
    ds = SyntheticDataset(cfg.vocab_size, cfg.seq_length, 8 * steps, seed=1)
    print(f"--- sched_total={total_sched}")
    for i in range(steps):
        rows = [ds[8 * i + j] for j in range(8)]
        batch = {k: torch.stack([r[k] for r in rows]) for k in rows[0]}
        t.engine.set_sync(True)
        out = t.train_step(batch)
        t.optimizer_step()
        print(i, "loss", round(float(out["ce_loss"].detach()), 3),
              "lr", f"{t.get_lr():.2e}",
              "gn", round(t.optimizer.last_grad_norm(), 3), flush=True)

run(10000)
run(21)
