"""Fused decoder vs the round-1 graph decoder on the b1 preset (dense at
decode: MoD layers run dense for single tokens)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from luminaai_amd.config import ConfigPresets
from luminaai_amd.models import DeepSeekTransformer, config_to_deepseek_config
from luminaai_amd.inference.fused_decode import FusedDecoder, can_fuse_decode

preset = sys.argv[1] if len(sys.argv) > 1 else "b1"
cfg = ConfigPresets.get(preset)
mcfg = config_to_deepseek_config(cfg)
torch.manual_seed(0)
with torch.device("cuda"):
    model = DeepSeekTransformer(mcfg)
model = model.to(torch.bfloat16).eval()
print(preset, "can_fuse:", can_fuse_decode(model))

dec = FusedDecoder(model, 2048)
prompt = torch.randint(1, 1000, (1, 64), device="cuda")

def measure(label, use_graph):
    dec.reset()
    logits = dec.prefill(prompt)
    if use_graph:
        dec.capture()
    tok = logits.argmax(-1).view(1)
    def run(n):
        nonlocal tok
        for _ in range(n):
            lg = dec.step(tok)
            tok = lg.float().argmax().view(1)
    run(10)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    run(100)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 100
    print(f"{label}: {dt*1e3:.2f} ms/token = {1/dt:.0f} tok/s")

measure("fused eager", False)
measure("fused graph", True)
measure("fused graph (2nd)", True)
