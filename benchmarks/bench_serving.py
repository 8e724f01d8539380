#!/usr/bin/env python3
"""Serving throughput microbench: continuous batching vs sequential decode.

Synthetic prompts through the flagship model family (dense by default —
greedy equivalence lets the two paths be compared token-for-token).

    python benchmarks/bench_serving.py [--requests 16] [--max-batch 8]
        [--prompt-len 64] [--max-new 32] [--hidden 512] [--layers 8]

Prints one JSON line with tokens/s for both paths and the speedup.
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--requests", type=int, default=16)
    ap.add_argument("--max-batch", type=int, default=8)
    ap.add_argument("--prompt-len", type=int, default=64)
    ap.add_argument("--max-new", type=int, default=32)
    ap.add_argument("--hidden", type=int, default=512)
    ap.add_argument("--layers", type=int, default=8)
    ap.add_argument("--vocab", type=int, default=4096)
    args = ap.parse_args()

    from luminaai_amd.data.tokenizer import ConversationTokenizer
    from luminaai_amd.inference.continuous import ContinuousBatchingEngine
    from luminaai_amd.inference.engine import (GenerationConfig,
                                               GenerationEngine)
    from luminaai_amd.models import DeepSeekTransformer
    from luminaai_amd.models.transformer import DeepSeekConfig

    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
    torch.manual_seed(0)
    model = DeepSeekTransformer(DeepSeekConfig(
        vocab_size=args.vocab, hidden_size=args.hidden,
        num_layers=args.layers, num_heads=args.hidden // 64,
        num_kv_heads=max(1, args.hidden // 256), seq_length=2048,
        use_moe=False, use_mod=False)).to(device, dtype).eval()
    tok = ConversationTokenizer()
    cfg = GenerationConfig(max_new_tokens=args.max_new, temperature=0.0,
                           stop_token_ids=[-1], max_context=2048)
    torch.manual_seed(7)
    prompts = [torch.randint(3, args.vocab, (args.prompt_len,)).tolist()
               for _ in range(args.requests)]

    def sync():
        if device.type == "cuda":
            torch.cuda.synchronize()

    # sequential baseline
    eng = GenerationEngine(model, tok, device)
    eng.generate(prompts[0], cfg)            # warmup
    sync()
    t0 = time.perf_counter()
    seq_out = [eng.generate(p, cfg) for p in prompts]
    sync()
    t_seq = time.perf_counter() - t0
    n_seq = sum(len(o) for o in seq_out)

    # continuous batching
    ceng = ContinuousBatchingEngine(model, tok, max_batch=args.max_batch,
                                    max_len=args.prompt_len + args.max_new + 4,
                                    device=device)
    sync()
    t0 = time.perf_counter()
    cont_out = ceng.run_to_completion(prompts, cfg)
    sync()
    t_cont = time.perf_counter() - t0
    n_cont = sum(len(o) for o in cont_out)
    assert cont_out == seq_out, "continuous output diverged from sequential"

    print(json.dumps({
        "metric": "serving_tokens_per_sec",
        "sequential_tok_s": round(n_seq / t_seq, 1),
        "continuous_tok_s": round(n_cont / t_cont, 1),
        "speedup": round(t_seq / t_cont, 2),
        "requests": args.requests, "max_batch": args.max_batch,
        "prompt_len": args.prompt_len, "max_new": args.max_new,
        "device": str(device), "dtype": str(dtype).split(".")[-1],
    }))


if __name__ == "__main__":
    main()
