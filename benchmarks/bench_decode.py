#!/usr/bin/env python3
"""Decoding throughput: KV-cached incremental generation tokens/s.

The reference decoded with a full re-forward per token (Chat.py:355-465);
this measures the fixed engine (prefill once + single-token steps on a
static preallocated cache)."""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from luminaai_amd.config import ConfigPresets  # noqa: E402
from luminaai_amd.data.tokenizer import ConversationTokenizer  # noqa: E402
from luminaai_amd.inference import GenerationConfig, GenerationEngine  # noqa: E402
from luminaai_amd.models import DeepSeekTransformer, config_to_deepseek_config  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--preset", default="b1")
    ap.add_argument("--new-tokens", type=int, default=128)
    ap.add_argument("--prompt-len", type=int, default=256)
    ap.add_argument("--json", default=None)
    args = ap.parse_args()

    dev = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    cfg = ConfigPresets.get(args.preset)
    torch.manual_seed(0)
    with torch.device(dev):
        model = DeepSeekTransformer(config_to_deepseek_config(cfg))
    model = model.to(torch.bfloat16 if dev.type == "cuda" else torch.float32)
    model.eval()
    tok = ConversationTokenizer(max_length=cfg.seq_length)
    eng = GenerationEngine(model, tok, dev)

    prompt = list(torch.randint(1, 256, (args.prompt_len,)).tolist())
    gcfg = GenerationConfig(max_new_tokens=args.new_tokens, temperature=0.8,
                            max_context=cfg.seq_length,
                            stop_token_ids=[-1])  # random model: don't stop
    # warmup
    eng.generate(prompt, GenerationConfig(max_new_tokens=8, temperature=0.8,
                                          max_context=cfg.seq_length,
                                          stop_token_ids=[-1]))
    if dev.type == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    out = eng.generate(prompt, gcfg)
    if dev.type == "cuda":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    res = {
        "metric": "decode_tokens_per_sec",
        "value": round(len(out) / dt, 2),
        "generated": len(out),
        "prompt_len": args.prompt_len,
        "preset": args.preset,
        "ms_per_token": round(dt / max(len(out), 1) * 1e3, 2),
        "device": torch.cuda.get_device_name(0) if dev.type == "cuda" else "cpu",
    }
    print(json.dumps(res))
    if args.json:
        with open(args.json, "w") as f:
            json.dump(res, f)


if __name__ == "__main__":
    main()
