"""Real-data convergence run (round-2 VERDICT item 7).

Trains the debug_200m preset (MoD) on REAL text -- the Python source and
docs present in the container image (no network; the tokenizer runs its
offline byte-level fallback) -- through the actual data pipeline
(BaseTrainingDataset -> Trainer), and logs a loss-vs-tokens curve plus a
held-out eval.  Artifacts: gpurun_out/convergence.csv + summary line.

Usage: python scripts/convergence_run.py [--minutes 8] [--preset debug_200m]
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from luminaai_amd.config import ConfigPresets
from luminaai_amd.data.dataset import BaseTrainingDataset, create_dataloader
from luminaai_amd.data.tokenizer import ConversationTokenizer


def build_corpus(path, cap_mb=24):
    """Concatenate real text from the image: python stdlib + installed
    package sources + any docs. Deterministic order."""
    roots = ["/usr/lib/python3.10", "/usr/local/lib/python3.10/dist-packages/numpy",
             "/usr/local/lib/python3.10/dist-packages/pandas",
             "/usr/local/lib/python3.10/dist-packages/torch/nn",
             "/usr/local/lib/python3.10/dist-packages/torch/distributed"]
    cap = cap_mb * 1024 * 1024
    n = 0
    with open(path, "w", encoding="utf-8") as out:
        for root in roots:
            for dirp, dirs, files in os.walk(root):
                dirs.sort()
                for f in sorted(files):
                    if not f.endswith((".py", ".txt", ".rst", ".md")):
                        continue
                    try:
                        txt = open(os.path.join(dirp, f), encoding="utf-8",
                                   errors="ignore").read()
                    except OSError:
                        continue
                    if len(txt) < 256:
                        continue
                    out.write(txt + "\n\n")
                    n += len(txt)
                    if n > cap:
                        return n
    return n


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--minutes", type=float, default=8.0)
    ap.add_argument("--preset", default="debug_200m")
    ap.add_argument("--seq-len", type=int, default=1024)
    args = ap.parse_args()

    corpus = "/tmp/corpus.txt"
    nbytes = build_corpus(corpus)
    print(f"corpus: {nbytes/1e6:.1f} MB of real text", flush=True)

    cfg = ConfigPresets.get(args.preset)
    tok = ConversationTokenizer(max_length=args.seq_len)
    cfg.vocab_size = ((tok.vocab_size + 127) // 128) * 128
    cfg.seq_length = args.seq_len
    cfg.micro_batch_size = 16
    cfg.gradient_accumulation_steps = 1
    cfg.num_workers = 0
    cfg.zero_stage = 0
    cfg.learning_rate = 3e-4
    cfg.warmup_steps = 100
    cfg.eval_every_n_batches = 0
    cfg.save_every_n_batches = 0
    cfg.gradient_checkpointing = False

    ds = BaseTrainingDataset(corpus, tok, cfg.seq_length)
    n_eval = max(8, len(ds) // 50)
    train_idx = list(range(len(ds) - n_eval))
    eval_idx = list(range(len(ds) - n_eval, len(ds)))
    train = torch.utils.data.Subset(ds, train_idx)
    evals = torch.utils.data.Subset(ds, eval_idx)
    print(f"dataset: {len(ds)} chunks of {cfg.seq_length} "
          f"({len(ds)*cfg.seq_length/1e6:.1f}M tokens), {n_eval} held out",
          flush=True)

    from luminaai_amd.models import DeepSeekTransformer, config_to_deepseek_config
    from luminaai_amd.training import Trainer
    torch.manual_seed(0)
    model = DeepSeekTransformer(config_to_deepseek_config(cfg))
    trainer = Trainer(model, tok, cfg)
    trainer._setup_scheduler(5000)

    loader = create_dataloader(train, cfg, shuffle=True)
    eval_loader = create_dataloader(evals, cfg, shuffle=False)

    budget_s = args.minutes * 60
    t0 = time.time()
    rows = []
    step = 0
    os.makedirs("gpurun_out", exist_ok=True)
    done = False
    while not done:
        for batch in loader:
            trainer.engine.set_sync(True)
            out = trainer.train_step(batch)
            trainer.optimizer_step()
            step += 1
            if step % 20 == 0:
                loss = float(out["ce_loss"])
                toks = step * cfg.micro_batch_size * cfg.seq_length
                dt = time.time() - t0
                rows.append((step, toks, loss, dt))
                print(f"step {step} tokens {toks/1e6:.1f}M loss {loss:.4f} "
                      f"({toks/dt:.0f} tok/s)", flush=True)
            if time.time() - t0 > budget_s:
                done = True
                break

    ev = trainer.evaluate(eval_loader)
    with open("gpurun_out/convergence.csv", "w") as f:
        f.write("step,tokens,loss,seconds\n")
        for r in rows:
            f.write(f"{r[0]},{r[1]},{r[2]:.5f},{r[3]:.1f}\n")
    print(f"FINAL: steps={step} eval_loss={ev['loss']:.4f} "
          f"eval_ppl={ev.get('perplexity', float('nan')):.2f} "
          f"first_loss={rows[0][2]:.3f} last_loss={rows[-1][2]:.3f}",
          flush=True)


if __name__ == "__main__":
    main()
