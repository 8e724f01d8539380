"""Compare attention paths on the b1 shape: AOTriton flash vs math/GEMM
composition, fwd and fwd+bwd."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F

B, Hq, Hkv, S, D = 16, 12, 4, 2048, 159
dt = torch.bfloat16
q = torch.randn(B, Hq, S, D, device="cuda", dtype=dt, requires_grad=True)
k = torch.randn(B, Hkv, S, D, device="cuda", dtype=dt, requires_grad=True)
v = torch.randn(B, Hkv, S, D, device="cuda", dtype=dt, requires_grad=True)

def t(fn, iters=10, warmup=3):
    for _ in range(warmup): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3

def fwd(backend):
    from torch.nn.attention import sdpa_kernel, SDPBackend
    with sdpa_kernel(backend):
        return F.scaled_dot_product_attention(q, k, v, is_causal=True,
                                              enable_gqa=True)

def fwdbwd(backend):
    from torch.nn.attention import sdpa_kernel, SDPBackend
    q.grad = k.grad = v.grad = None
    with sdpa_kernel(backend):
        o = F.scaled_dot_product_attention(q, k, v, is_causal=True,
                                           enable_gqa=True)
    o.backward(torch.ones_like(o))

from torch.nn.attention import SDPBackend
for name, be in [("flash", SDPBackend.FLASH_ATTENTION),
                 ("efficient", SDPBackend.EFFICIENT_ATTENTION),
                 ("math", SDPBackend.MATH)]:
    try:
        f = t(lambda: fwd(be))
        fb = t(lambda: fwdbwd(be), iters=5)
        print(f"{name:10s} fwd {f:8.3f} ms   fwd+bwd {fb:8.3f} ms")
    except Exception as e:
        print(f"{name:10s} unavailable: {str(e)[:80]}")

# explicit GEMM composition (what a custom chunked backward would cost):
def explicit():
    kk = k.repeat_interleave(Hq // Hkv, dim=1)
    vv = v.repeat_interleave(Hq // Hkv, dim=1)
    s = q @ kk.transpose(-1, -2) / (D ** 0.5)
    mask = torch.ones(S, S, device="cuda", dtype=torch.bool).tril()
    s = s.masked_fill(~mask, float("-inf"))
    p = s.float().softmax(-1).to(dt)
    return p @ vv
print(f"explicit-fwd {t(explicit, iters=5):8.3f} ms")
