import os, sys, time, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from luminaai_amd.config import ConfigPresets
from luminaai_amd.data.tokenizer import ConversationTokenizer
from luminaai_amd.inference import GenerationConfig, GenerationEngine
from luminaai_amd.models import DeepSeekTransformer, config_to_deepseek_config
from luminaai_amd.inference.graph_decode import GraphedDecoder

dev = torch.device("cuda")
cfg = ConfigPresets.get("b1")
torch.manual_seed(0)
with torch.device(dev):
    m = DeepSeekTransformer(config_to_deepseek_config(cfg))
m = m.to(torch.bfloat16).eval()
tok = ConversationTokenizer(max_length=2048)
eng = GenerationEngine(m, tok, dev)

prompt = list(torch.randint(1, 256, (256,)).tolist())
for label, ug, temp in [("eager-sampled", False, 0.8), ("graph-sampled", True, 0.8),
                        ("eager-greedy", False, 0.0), ("graph-greedy", True, 0.0)]:
    g = GenerationConfig(max_new_tokens=48, temperature=temp, max_context=2048,
                         stop_token_ids=[-1])
    eng.generate(prompt, GenerationConfig(max_new_tokens=4, temperature=temp,
                                          max_context=2048, stop_token_ids=[-1]),
                 use_graph=ug)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    out = eng.generate(prompt, g, use_graph=ug)
    torch.cuda.synchronize(); dt = time.perf_counter() - t0
    print(f"{label}: {len(out)/dt:.1f} tok/s ({dt/max(len(out),1)*1e3:.2f} ms/tok)", flush=True)

# raw replay rate (no sampling):
dec = eng._graph_dec
tokk = torch.ones(1, 1, dtype=torch.long, device=dev)
dec.reset(); dec.prefill(torch.tensor([prompt], device=dev))
for _ in range(3): dec.step(tokk)
torch.cuda.synchronize(); t0 = time.perf_counter()
for _ in range(48): dec.step(tokk)
torch.cuda.synchronize()
print(f"raw graph replay: {(time.perf_counter()-t0)/48*1e3:.2f} ms/tok", flush=True)

# MoE capture error detail
from luminaai_amd.models.transformer import DeepSeekConfig
mcfg = DeepSeekConfig(vocab_size=512, hidden_size=128, num_layers=2,
                      num_heads=4, num_kv_heads=2, intermediate_size=256,
                      seq_length=64, use_moe=True, num_experts=4,
                      moe_top_k=2, use_mod=False, routing_noise_std=0.0)
mm = DeepSeekTransformer(mcfg).to(dev, torch.bfloat16).eval()
d2 = GraphedDecoder(mm, max_context=32)
d2.prefill(torch.randint(1, 512, (1, 8), device=dev))
try:
    d2.step(torch.ones(1, 1, dtype=torch.long, device=dev))
    print("moe graph step OK")
except Exception as e:
    print("MoE capture error:", type(e).__name__, str(e)[:500], flush=True)
