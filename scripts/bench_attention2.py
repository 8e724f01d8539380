"""Hand-written CDNA4 flash attention vs AOTriton SDPA on the b1 geometry.

Reports ms and effective TF for fwd and fwd+bwd on the standard flash-FLOP
accounting (fwd 2 matmuls, bwd 5 -> causal halves the work).
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F

from luminaai_amd.ops import flash_attention

B, Hq, Hkv, S, D = 16, 12, 4, 2048, 159
dt = torch.bfloat16
torch.manual_seed(0)
# model layout [B,S,H,D]
q = torch.randn(B, S, Hq, D, device="cuda", dtype=dt, requires_grad=True)
k = torch.randn(B, S, Hkv, D, device="cuda", dtype=dt, requires_grad=True)
v = torch.randn(B, S, Hkv, D, device="cuda", dtype=dt, requires_grad=True)
go = torch.randn(B, S, Hq, D, device="cuda", dtype=dt)
scale = D ** -0.5

# flash-FLOP accounting (causal: x0.5)
fwd_flop = 2 * 2 * B * Hq * S * S * D * 0.5
bwd_flop = 5 * 2 * B * Hq * S * S * D * 0.5


def t(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def own_fwd():
    return flash_attention(q, k, v, scale)


def own_fwdbwd():
    q.grad = k.grad = v.grad = None
    o = flash_attention(q, k, v, scale)
    o.backward(go)


def sdpa_fwd():
    return F.scaled_dot_product_attention(
        q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
        is_causal=True, enable_gqa=True)


def sdpa_fwdbwd():
    q.grad = k.grad = v.grad = None
    o = F.scaled_dot_product_attention(
        q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
        is_causal=True, enable_gqa=True)
    o.backward(go.transpose(1, 2))


for name, ffwd, ffb in [("own", own_fwd, own_fwdbwd),
                        ("sdpa", sdpa_fwd, sdpa_fwdbwd)]:
    f = t(ffwd)
    fb = t(ffb)
    bwd_ms = fb - f
    print(f"{name}: fwd {f:.3f} ms ({fwd_flop/f/1e9:.0f} TF)  "
          f"fwd+bwd {fb:.3f} ms  bwd {bwd_ms:.3f} ms "
          f"({bwd_flop/bwd_ms/1e9:.0f} TF)")
