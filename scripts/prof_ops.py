"""Op-level torch.profiler pass over one b1_moe training step: attributes
the at::native copy/add/fill glue kernels (12% of the r2-final profile)
to their source aten ops."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from luminaai_amd.config import ConfigPresets
from luminaai_amd.data.dataset import SyntheticDataset
from luminaai_amd.data.tokenizer import ConversationTokenizer
from luminaai_amd.models import DeepSeekTransformer, config_to_deepseek_config
from luminaai_amd.training import Trainer

cfg = ConfigPresets.get("b1_moe")
cfg.micro_batch_size = 16
cfg.gradient_accumulation_steps = 2
cfg.num_workers = 0
cfg.gradient_checkpointing = False   # match bench.py (preset default is True)
cfg.eval_every_n_batches = 0
cfg.save_every_n_batches = 0
device = torch.device("cuda")
torch.manual_seed(1)
with torch.device(device):
    model = DeepSeekTransformer(config_to_deepseek_config(cfg))
trainer = Trainer(model, ConversationTokenizer(max_length=cfg.seq_length), cfg)
trainer._setup_scheduler(30)
ds = SyntheticDataset(cfg.vocab_size, cfg.seq_length, 16 * 2 * 4, seed=1)
batches = [ds[i] for i in range(len(ds))]

def step(i):
    for a in range(2):
        trainer.engine.set_sync(a == 1)
        rows = [batches[(i * 2 + a) * 16 + j] for j in range(16)]
        batch = {k: torch.stack([r[k] for r in rows]) for k in rows[0]}
        trainer.train_step(batch)
    trainer.optimizer_step()

step(0)
torch.cuda.synchronize()
from torch.profiler import profile, ProfilerActivity
with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
             record_shapes=True) as prof:
    step(1)
    torch.cuda.synchronize()
print(prof.key_averages(group_by_input_shape=True)
      .table(sort_by="self_cuda_time_total", row_limit=30))
